// backend.js — BackendService equivalent (ref: kubeflow-common-lib
// backend.service.ts:19-70): parses the crud_backend envelope
// {status, success, user, ...} / {success:false, log, status}, carries the
// double-submit CSRF token (XSRF-TOKEN cookie -> X-XSRF-TOKEN header) and
// funnels failures to the snackbar.
import { snack } from "./components.js";

function csrfToken() {
  const m = document.cookie.match(/(?:^|;\s*)XSRF-TOKEN=([^;]+)/);
  return m ? decodeURIComponent(m[1]) : null;
}

export async function api(method, path, body) {
  const headers = { "Content-Type": "application/json" };
  const tok = csrfToken();
  if (tok) headers["X-XSRF-TOKEN"] = tok;
  let resp;
  try {
    resp = await fetch(path, {
      method,
      headers,
      body: body === undefined ? undefined : JSON.stringify(body),
    });
  } catch (e) {
    snack(`network error: ${e.message}`, "error");
    throw e;
  }
  let data = {};
  try {
    data = await resp.json();
  } catch {
    /* non-JSON (shouldn't happen through the envelope) */
  }
  if (data.success === false || !resp.ok) {
    const msg = data.log || `${resp.status} ${resp.statusText}`;
    snack(msg, "error");
    const err = new Error(msg);
    err.status = data.status || resp.status;
    throw err;
  }
  return data;
}

export const get = (p) => api("GET", p);
export const post = (p, b) => api("POST", p, b);
export const patch = (p, b) => api("PATCH", p, b);
export const del = (p) => api("DELETE", p);

// current namespace selection, persisted like namespace-selector.js:106-170
const NS_KEY = "kf.selectedNamespace";
export function selectedNamespace() {
  return localStorage.getItem(NS_KEY) || "default";
}
export function selectNamespace(ns) {
  localStorage.setItem(NS_KEY, ns);
  window.dispatchEvent(new CustomEvent("kf-namespace", { detail: ns }));
}
