from .bindings import binding_name, BindingClient

__all__ = ["binding_name", "BindingClient"]
