// components.js — shared UI: snackbar, status icons, confirm dialog,
// generic resource table (the kubeflow-common-lib component set:
// resource-table, status icons, confirm-dialog, snack-bar).

export function el(tag, attrs = {}, ...children) {
  const n = document.createElement(tag);
  for (const [k, v] of Object.entries(attrs)) {
    if (k === "class") n.className = v;
    else if (k.startsWith("on")) n.addEventListener(k.slice(2), v);
    else if (v !== undefined && v !== null) n.setAttribute(k, v);
  }
  for (const c of children.flat()) {
    if (c === null || c === undefined) continue;
    n.append(c.nodeType ? c : document.createTextNode(String(c)));
  }
  return n;
}

export function snack(msg, kind = "info") {
  let host = document.getElementById("kf-snackbar");
  if (!host) {
    host = el("div", { id: "kf-snackbar" });
    document.body.append(host);
  }
  const item = el("div", { class: `snack snack-${kind}` }, msg);
  host.append(item);
  setTimeout(() => item.remove(), kind === "error" ? 8000 : 4000);
}

export function confirmDialog(message) {
  return new Promise((resolve) => {
    const overlay = el("div", { class: "kf-overlay" });
    const box = el(
      "div",
      { class: "kf-dialog" },
      el("p", {}, message),
      el(
        "div",
        { class: "kf-dialog-actions" },
        el("button", {
          class: "btn",
          onclick: () => {
            overlay.remove();
            resolve(false);
          },
        }, "Cancel"),
        el("button", {
          class: "btn btn-danger",
          "data-testid": "confirm-yes",
          onclick: () => {
            overlay.remove();
            resolve(true);
          },
        }, "Delete"),
      ),
    );
    overlay.append(box);
    document.body.append(overlay);
  });
}

// status from conditions[] — the derivation pattern of
// jupyter/apps/common/status.py:10-57 collapsed to the process model
export function statusOf(obj) {
  const conds = (obj.status && obj.status.conditions) || [];
  const by = {};
  for (const c of conds) by[c.type] = c;
  const stopped =
    obj.metadata.annotations &&
    obj.metadata.annotations["kubeflow-resource-stopped"];
  if (stopped) return { phase: "stopped", icon: "⏸", cls: "st-stopped" };
  if (by.Failed && by.Failed.status === "True")
    return { phase: "error", icon: "✖", cls: "st-error", msg: by.Failed.message };
  if (by.Succeeded && by.Succeeded.status === "True")
    return { phase: "succeeded", icon: "✔", cls: "st-ok" };
  if (
    (by.Ready && by.Ready.status === "True") ||
    (by.Running && by.Running.status === "True")
  )
    return { phase: "ready", icon: "✔", cls: "st-ok" };
  return { phase: "waiting", icon: "⟳", cls: "st-warn" };
}

export function age(obj) {
  const t = obj.metadata.creationTimestamp;
  if (!t) return "";
  const s = Math.max(0, (Date.now() - new Date(t).getTime()) / 1000);
  if (s < 90) return `${Math.round(s)}s`;
  if (s < 5400) return `${Math.round(s / 60)}m`;
  if (s < 129600) return `${Math.round(s / 3600)}h`;
  return `${Math.round(s / 86400)}d`;
}

// generic resource table: columns = [{title, render(obj)}], actions =
// [{label, run(obj), cls?, when?(obj)}]
export function resourceTable({ items, columns, actions = [], empty }) {
  if (!items.length)
    return el("p", { class: "kf-empty" }, empty || "No resources found.");
  const head = el(
    "tr",
    {},
    columns.map((c) => el("th", {}, c.title)),
    actions.length ? el("th", {}, "") : null,
  );
  const rows = items.map((o) =>
    el(
      "tr",
      { "data-name": o.metadata.name },
      columns.map((c) => el("td", {}, c.render(o))),
      actions.length
        ? el(
            "td",
            { class: "kf-actions" },
            actions
              .filter((a) => !a.when || a.when(o))
              .map((a) =>
                el(
                  "button",
                  {
                    class: `btn btn-small ${a.cls || ""}`,
                    "data-action": a.label.toLowerCase(),
                    onclick: () => a.run(o),
                  },
                  a.label,
                ),
              ),
          )
        : null,
    ),
  );
  return el("table", { class: "kf-table" }, el("thead", {}, head), el("tbody", {}, rows));
}

export function statusCell(o) {
  const st = statusOf(o);
  return el(
    "span",
    { class: `st ${st.cls}`, title: st.msg || st.phase },
    `${st.icon} ${st.phase}`,
  );
}
