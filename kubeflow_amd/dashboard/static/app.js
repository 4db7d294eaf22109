// app.js — dashboard shell: namespace selector + hash router over the
// page views (the central-dashboard main-page equivalent; instead of
// iframing per-resource Angular apps — main-page.js:244-307 — the pages
// are ES-module views over the same BFF).
import { get, selectedNamespace, selectNamespace } from "./backend.js";
import { el } from "./components.js";
import {
  homePage,
  notebooksPage,
  notebookSpawnerPage,
  volumesPage,
  tensorboardsPage,
  jobsPage,
  jobLogsPage,
  servingPage,
  experimentsPage,
  pipelinesPage,
  activitiesPage,
  workgroupPage,
} from "./pages.js";

const ROUTES = [
  [/^#?\/?$/, () => homePage(), "Home"],
  [/^#\/notebooks\/new$/, () => notebookSpawnerPage()],
  [/^#\/notebooks$/, () => notebooksPage(), "Notebooks"],
  [/^#\/volumes$/, () => volumesPage(), "Volumes"],
  [/^#\/tensorboards$/, () => tensorboardsPage(), "Tensorboards"],
  [/^#\/jobs\/([^/]+)\/([^/]+)\/logs$/, (m) => jobLogsPage(m[1], m[2])],
  [/^#\/jobs$/, () => jobsPage(), "Jobs"],
  [/^#\/serving$/, () => servingPage(), "Serving"],
  [/^#\/experiments$/, () => experimentsPage(), "Experiments"],
  [/^#\/pipelines$/, () => pipelinesPage(), "Pipelines"],
  [/^#\/activities$/, () => activitiesPage(), "Activity"],
  [/^#\/workgroup$/, () => workgroupPage(), "Workgroup"],
];

let current = null;

function route() {
  const hash = location.hash || "#/";
  for (const [re, make] of ROUTES) {
    const m = hash.match(re);
    if (m) {
      if (current) current.stop();
      current = make(m);
      const main = document.getElementById("kf-main");
      main.replaceChildren(current.node);
      for (const a of document.querySelectorAll("#kf-nav a"))
        a.classList.toggle("active", a.getAttribute("href") === hash);
      return;
    }
  }
  location.hash = "#/";
}

async function boot() {
  const navItems = [
    ["#/", "Home"],
    ["#/notebooks", "Notebooks"],
    ["#/tensorboards", "Tensorboards"],
    ["#/volumes", "Volumes"],
    ["#/jobs", "Jobs"],
    ["#/serving", "Serving"],
    ["#/experiments", "Experiments"],
    ["#/pipelines", "Pipelines"],
    ["#/activities", "Activity"],
    ["#/workgroup", "Workgroup"],
  ];
  const nav = el(
    "nav",
    { id: "kf-nav" },
    navItems.map(([href, label]) => el("a", { href }, label)),
  );
  const nsSelect = el("select", {
    id: "kf-ns",
    onchange: (ev) => {
      selectNamespace(ev.target.value);
      route(); // re-render the current page in the new namespace
    },
  });
  const userSpan = el("span", { id: "kf-user" }, "…");
  document.body.prepend(
    el(
      "header",
      { id: "kf-header" },
      el("span", { class: "kf-logo" }, "kubeflow-amd"),
      el("label", { class: "kf-ns-label" }, "namespace ", nsSelect),
      userSpan,
    ),
  );
  const layout = el(
    "div",
    { id: "kf-layout" },
    nav,
    el("main", { id: "kf-main" }),
  );
  document.body.append(layout);

  try {
    const data = await get("/api/namespaces");
    const names = data.namespaces && data.namespaces.length
      ? data.namespaces
      : ["default"];
    if (!names.includes("default")) names.unshift("default");
    const sel = selectedNamespace();
    nsSelect.replaceChildren(
      ...names.map((n) =>
        el("option", { value: n, ...(n === sel ? { selected: "" } : {}) }, n),
      ),
    );
    userSpan.textContent = data.user || "";
  } catch {
    nsSelect.replaceChildren(el("option", { value: "default" }, "default"));
  }

  window.addEventListener("hashchange", route);
  route();
  document.body.dataset.ready = "1"; // e2e readiness marker
}

boot();
