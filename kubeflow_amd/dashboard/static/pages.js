// pages.js — page views over the BFF. Each page returns {node, stop()}.
// Lists poll with ExponentialBackoff and reset-on-change (the reference's
// UI reconcile pattern); mutations go through backend.js (envelope+CSRF).
import { api, get, post, patch, del, selectedNamespace } from "./backend.js";
import { ExponentialBackoff, changed } from "./polling.js";
import {
  el,
  snack,
  confirmDialog,
  resourceTable,
  statusCell,
  age,
} from "./components.js";

function listPage({ title, plural, columns, actions, createHref, extras }) {
  const ns = selectedNamespace();
  const body = el("div", {}, el("p", { class: "kf-empty" }, "Loading…"));
  let prev = null;
  const poller = new ExponentialBackoff(async () => {
    const data = await get(`/api/namespaces/${ns}/${plural}`);
    const items = data.items || data[plural] || [];
    if (changed(prev, items)) poller.reset();
    prev = items;
    body.replaceChildren(
      resourceTable({
        items,
        columns,
        actions,
        empty: `No ${title.toLowerCase()} in ${ns}.`,
      }),
    );
  });
  poller.start();
  const node = el(
    "div",
    {},
    el(
      "div",
      { class: "kf-page-head" },
      el("h2", {}, title),
      createHref
        ? el(
            "a",
            { class: "btn btn-primary", href: createHref, "data-testid": "new" },
            "+ New",
          )
        : null,
    ),
    extras || null,
    body,
  );
  return { node, stop: () => poller.stop() };
}

const nameCol = { title: "Name", render: (o) => o.metadata.name };
const statusCol = { title: "Status", render: (o) => statusCell(o) };
const ageCol = { title: "Age", render: (o) => age(o) };

function deleteAction(plural, label = "Delete") {
  return {
    label,
    cls: "btn-danger",
    run: async (o) => {
      const m = o.metadata;
      if (!(await confirmDialog(`Delete ${m.name}? This cannot be undone.`)))
        return;
      await del(`/api/namespaces/${m.namespace}/${plural}/${m.name}`);
      snack(`${m.name} deleted`);
    },
  };
}

// ----------------------------------------------------------- notebooks
export function notebooksPage() {
  const stopAction = {
    label: "Stop",
    when: (o) =>
      !(o.metadata.annotations || {})["kubeflow-resource-stopped"],
    run: async (o) => {
      await patch(
        `/api/namespaces/${o.metadata.namespace}/notebooks/${o.metadata.name}`,
        { stopped: true },
      );
      snack(`${o.metadata.name} stopping`);
    },
  };
  const startAction = {
    label: "Start",
    when: (o) =>
      !!(o.metadata.annotations || {})["kubeflow-resource-stopped"],
    run: async (o) => {
      await patch(
        `/api/namespaces/${o.metadata.namespace}/notebooks/${o.metadata.name}`,
        { stopped: false },
      );
      snack(`${o.metadata.name} starting`);
    },
  };
  return listPage({
    title: "Notebooks",
    plural: "notebooks",
    createHref: "#/notebooks/new",
    columns: [
      nameCol,
      statusCol,
      { title: "Image", render: (o) => o.spec.image || "" },
      {
        title: "GPUs",
        render: (o) => String(o.spec.gpus || 0),
      },
      {
        title: "Connect",
        render: (o) =>
          o.status && o.status.url
            ? el("a", { href: o.status.url, target: "_blank" }, "open")
            : "—",
      },
      ageCol,
    ],
    actions: [stopAction, startAction, deleteAction("notebooks")],
  });
}

// spawner form driven by /api/config's value/readOnly admin semantics
// (ref: jupyter form.py:121-336 + spawner_ui_config.yaml)
export function notebookSpawnerPage() {
  const ns = selectedNamespace();
  const node = el("div", {}, el("h2", {}, "New Notebook"), el("p", {}, "Loading config…"));
  (async () => {
    const data = await get("/api/config");
    const cfg = data.config || {};
    const field = (id, label, value, readOnly, type = "text") =>
      el(
        "label",
        { class: "kf-field" },
        el("span", {}, label),
        el("input", {
          id,
          type,
          value,
          ...(readOnly ? { disabled: "" } : {}),
        }),
      );
    const gpuCfg = (cfg.gpus && cfg.gpus.value) || {};
    const form = el(
      "form",
      {
        class: "kf-form",
        "data-testid": "spawner-form",
        onsubmit: async (ev) => {
          ev.preventDefault();
          const name = form.querySelector("#nb-name").value.trim();
          if (!name) return snack("name is required", "error");
          const gpus = parseInt(form.querySelector("#nb-gpus").value) || 0;
          try {
            await post(`/api/namespaces/${ns}/notebooks`, {
              name,
              image: form.querySelector("#nb-image").value,
              cpu: form.querySelector("#nb-cpu").value,
              memory: form.querySelector("#nb-mem").value,
              gpus,
            });
            snack(`notebook ${name} created`);
            location.hash = "#/notebooks";
          } catch {
            /* snack already shown */
          }
        },
      },
      field("nb-name", "Name", "", false),
      field("nb-image", "Image", (cfg.image && cfg.image.value) || "",
            cfg.image && cfg.image.readOnly),
      field("nb-cpu", "CPU", (cfg.cpu && cfg.cpu.value) || "2",
            cfg.cpu && cfg.cpu.readOnly),
      field("nb-mem", "Memory", (cfg.memory && cfg.memory.value) || "4Gi",
            cfg.memory && cfg.memory.readOnly),
      field(
        "nb-gpus",
        `GPUs (${(gpuCfg.vendors || [])
          .map((v) => v.uiName)
          .join("/") || "AMD"})`,
        gpuCfg.num === "none" ? "0" : gpuCfg.num || "0",
        cfg.gpus && cfg.gpus.readOnly,
        "number",
      ),
      el(
        "button",
        { class: "btn btn-primary", type: "submit", "data-testid": "spawn" },
        "Launch",
      ),
    );
    node.replaceChildren(el("h2", {}, "New Notebook"), form);
  })();
  return { node, stop: () => {} };
}

// ----------------------------------------------------------- volumes
export function volumesPage() {
  return listPage({
    title: "Volumes",
    plural: "pvcs",
    columns: [
      nameCol,
      {
        title: "Phase",
        render: (o) => (o.status && o.status.phase) || "Pending",
      },
      {
        title: "Size",
        render: (o) =>
          (o.spec.resources &&
            o.spec.resources.requests &&
            o.spec.resources.requests.storage) ||
          "",
      },
      ageCol,
    ],
    actions: [deleteAction("pvcs")],
  });
}

// ----------------------------------------------------------- tensorboards
export function tensorboardsPage() {
  return listPage({
    title: "Tensorboards",
    plural: "tensorboards",
    columns: [
      nameCol,
      statusCol,
      { title: "Logs path", render: (o) => o.spec.logspath || "" },
      ageCol,
    ],
    actions: [deleteAction("tensorboards")],
  });
}

// ----------------------------------------------------------- jobs
export function jobsPage() {
  const ns = selectedNamespace();
  const submit = el(
    "form",
    {
      class: "kf-inline-form",
      "data-testid": "job-form",
      onsubmit: async (ev) => {
        ev.preventDefault();
        const name = submit.querySelector("#job-name").value.trim();
        const replicas =
          parseInt(submit.querySelector("#job-replicas").value) || 1;
        const model = submit.querySelector("#job-model").value || "mnist-mlp";
        if (!name) return snack("name is required", "error");
        try {
          await post(`/api/namespaces/${ns}/pytorchjobs`, {
            apiVersion: "kubeflow.org/v1",
            kind: "PyTorchJob",
            metadata: { name, namespace: ns },
            spec: {
              pytorchReplicaSpecs: {
                Worker: {
                  replicas,
                  restartPolicy: "Never",
                  template: { model, steps: 5, gpus_per_replica: 0 },
                },
              },
            },
          });
          snack(`PyTorchJob ${name} submitted`);
        } catch {
          /* snack shown */
        }
      },
    },
    el("input", { id: "job-name", placeholder: "job name" }),
    el("input", { id: "job-model", placeholder: "model", value: "mnist-mlp" }),
    el("input", {
      id: "job-replicas",
      type: "number",
      value: "1",
      min: "1",
      max: "8",
    }),
    el(
      "button",
      { class: "btn btn-primary", type: "submit", "data-testid": "submit-job" },
      "Submit PyTorchJob",
    ),
  );
  return listPage({
    title: "Training Jobs",
    plural: "pytorchjobs",
    extras: submit,
    columns: [
      nameCol,
      statusCol,
      {
        title: "Replicas",
        render: (o) => {
          const w =
            (o.spec.pytorchReplicaSpecs && o.spec.pytorchReplicaSpecs.Worker) ||
            {};
          return String(w.replicas || 1);
        },
      },
      {
        title: "Logs",
        render: (o) =>
          el(
            "a",
            {
              href: `#/jobs/${o.metadata.namespace}/${o.metadata.name}/logs`,
            },
            "view",
          ),
      },
      ageCol,
    ],
    actions: [deleteAction("pytorchjobs")],
  });
}

export function jobLogsPage(ns, name) {
  const pre = el("pre", { class: "kf-logs" }, "Loading…");
  const poller = new ExponentialBackoff(async () => {
    const data = await get(
      `/api/namespaces/${ns}/pytorchjobs/${name}/logs?tail=200`,
    );
    pre.textContent = (data.logs || []).join("\n") || "(no output yet)";
  });
  poller.start();
  return {
    node: el(
      "div",
      {},
      el("h2", {}, `Logs — ${ns}/${name}`),
      el("a", { href: "#/jobs" }, "← back to jobs"),
      pre,
    ),
    stop: () => poller.stop(),
  };
}

// ----------------------------------------------------------- serving
export function servingPage() {
  return listPage({
    title: "Model Serving",
    plural: "inferenceservices",
    columns: [
      nameCol,
      statusCol,
      {
        title: "Model",
        render: (o) => (o.spec.predictor && o.spec.predictor.model) || "",
      },
      {
        title: "Storage",
        render: (o) =>
          (o.spec.predictor && o.spec.predictor.storageUri) || "random-init",
      },
      {
        title: "URL",
        render: (o) => {
          const c = ((o.status && o.status.conditions) || []).find(
            (x) => x.type === "Ready" && x.status === "True",
          );
          return c && c.message && c.message.startsWith("http")
            ? el("a", { href: c.message, target: "_blank" }, c.message)
            : "—";
        },
      },
      ageCol,
    ],
    actions: [deleteAction("inferenceservices")],
  });
}

// ----------------------------------------------------------- experiments
export function experimentsPage() {
  return listPage({
    title: "Experiments (AutoML)",
    plural: "experiments",
    columns: [
      nameCol,
      statusCol,
      {
        title: "Trials",
        render: (o) => String((o.status && o.status.trials) || 0),
      },
      {
        title: "Best",
        render: (o) => {
          const b = o.status && o.status.currentOptimalTrial;
          if (!b || !b.observation) return "—";
          const m = (b.observation.metrics || [])[0];
          return m ? `${m.name}=${Number(m.latest).toFixed(4)}` : "—";
        },
      },
      ageCol,
    ],
    actions: [deleteAction("experiments")],
  });
}

// ----------------------------------------------------------- pipelines
export function pipelinesPage() {
  return listPage({
    title: "Pipelines",
    plural: "pipelineruns",
    columns: [
      nameCol,
      statusCol,
      {
        title: "Tasks",
        render: (o) => {
          const ts = (o.status && o.status.taskStates) || {};
          return Object.entries(ts)
            .map(([k, v]) => `${k}:${v}`)
            .join(" ");
        },
      },
      ageCol,
    ],
    actions: [deleteAction("pipelineruns")],
  });
}

// ----------------------------------------------------------- activities
export function activitiesPage() {
  const ns = selectedNamespace();
  const body = el("div", {}, "Loading…");
  const poller = new ExponentialBackoff(async () => {
    const data = await get(`/api/activities/${ns}`);
    const evs = data.activities || [];
    body.replaceChildren(
      el(
        "table",
        { class: "kf-table" },
        el(
          "thead",
          {},
          el("tr", {}, el("th", {}, "Time"), el("th", {}, "Type"),
             el("th", {}, "Reason"), el("th", {}, "Object"),
             el("th", {}, "Message")),
        ),
        el(
          "tbody",
          {},
          evs.slice(0, 50).map((e) =>
            el(
              "tr",
              { class: e.type === "Warning" ? "row-warn" : "" },
              el("td", {}, e.lastTimestamp || ""),
              el("td", {}, e.type || "Normal"),
              el("td", {}, e.reason || ""),
              el(
                "td",
                {},
                `${(e.involvedObject && e.involvedObject.kind) || ""}/${
                  (e.involvedObject && e.involvedObject.name) || ""
                }`,
              ),
              el("td", {}, e.message || ""),
            ),
          ),
        ),
      ),
    );
  });
  poller.start();
  return {
    node: el("div", {}, el("h2", {}, `Activity — ${ns}`), body),
    stop: () => poller.stop(),
  };
}

// ----------------------------------------------------------- home
export function homePage() {
  const body = el("div", { class: "kf-cards" });
  const poller = new ExponentialBackoff(async () => {
    const [links, metrics] = await Promise.all([
      get("/api/dashboard-links"),
      get("/api/metrics/node"),
    ]);
    const util = metrics.scheduler || {};
    const gpus = metrics.gpus || [];
    body.replaceChildren(
      el(
        "div",
        { class: "kf-card" },
        el("h3", {}, "Cluster"),
        el(
          "p",
          {},
          `GPUs: ${util.total_gpus || 0} (busy ${util.exclusive_busy || 0}) · jobs: ${util.jobs || 0}`,
        ),
        gpus.length
          ? el(
              "div",
              {},
              gpus.map((g) => {
                const pct = g.hbm_total
                  ? Math.round((100 * g.hbm_used) / g.hbm_total)
                  : 0;
                return el(
                  "div",
                  { class: "kf-bar" },
                  el("span", {}, `GPU ${g.index} HBM ${pct}%`),
                  el(
                    "div",
                    { class: "bar-track" },
                    el("div", { class: "bar-fill", style: `width:${pct}%` }),
                  ),
                );
              }),
            )
          : el("p", { class: "kf-empty" }, "No GPUs visible (CPU mode)."),
      ),
      el(
        "div",
        { class: "kf-card" },
        el("h3", {}, "Quick links"),
        el(
          "ul",
          {},
          (links.quickLinks || []).map((l) =>
            el("li", {}, el("a", { href: "#/notebooks/new" }, l.text)),
          ),
          el("li", {}, el("a", { href: "#/jobs" }, "Submit a PyTorchJob")),
        ),
      ),
    );
  }, { interval: 4000, maxInterval: 16000 });
  poller.start();
  return {
    node: el("div", {}, el("h2", {}, "Home"), body),
    stop: () => poller.stop(),
  };
}

// ----------------------------------------------------------- workgroup
export function workgroupPage() {
  const node = el("div", {}, el("h2", {}, "Workgroup"), el("p", {}, "Loading…"));
  (async () => {
    const info = await get("/api/workgroup/exists");
    if (info.hasWorkgroup) {
      const env = await get("/api/workgroup/env-info");
      node.replaceChildren(
        el("h2", {}, "Workgroup"),
        el("p", {}, `Signed in as ${env.user || info.user}`),
        el(
          "table",
          { class: "kf-table" },
          el("thead", {}, el("tr", {}, el("th", {}, "Namespace"), el("th", {}, "Role"))),
          el(
            "tbody",
            {},
            (env.namespaces || []).map((n) =>
              el("tr", {}, el("td", {}, n.namespace), el("td", {}, n.role)),
            ),
          ),
        ),
      );
      return;
    }
    // registration flow (ref: registration-page.js:91-112)
    const form = el(
      "form",
      {
        class: "kf-form",
        onsubmit: async (ev) => {
          ev.preventDefault();
          const nsName = form.querySelector("#wg-ns").value.trim();
          await post("/api/workgroup/create", { namespace: nsName });
          snack(`namespace ${nsName} created`);
          location.reload();
        },
      },
      el(
        "label",
        { class: "kf-field" },
        el("span", {}, "Namespace"),
        el("input", { id: "wg-ns", placeholder: "my-team" }),
      ),
      el("button", { class: "btn btn-primary", type: "submit" }, "Create workgroup"),
    );
    node.replaceChildren(
      el("h2", {}, "Welcome"),
      el("p", {}, "You have no workgroup yet — create one to get a namespace."),
      form,
    );
  })();
  return { node, stop: () => {} };
}
