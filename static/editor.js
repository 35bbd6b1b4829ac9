// Config editor logic (load/save via /v1/config/*, agent-config exports).
// Functional parity with the reference's editor.js:169-334: syntax-
// highlighted JSON-with-comments editing (self-hosted jsonc-editor.js, no
// CDN CodeMirror), inline validation errors at the offending line, theme
// picker persisted in localStorage, tabs, and the OpenCode/Copilot
// exports with the includefallback toggle.

(function () {
  "use strict";

  // ---- theme (4 themes; persisted like the reference's localStorage
  // codeMirrorTheme) ----
  const root = document.documentElement;
  const themeSel = document.getElementById("themeSelector");
  function applyTheme(t) {
    root.dataset.theme = t;
    localStorage.setItem("gw-theme", t);
    if (themeSel.value !== t) themeSel.value = t;
  }
  applyTheme(localStorage.getItem("gw-theme") || "dark");
  themeSel.addEventListener("change", () => applyTheme(themeSel.value));

  // ---- tabs ----
  document.querySelectorAll(".tab").forEach((btn) => {
    btn.onclick = () => {
      document.querySelectorAll(".tab").forEach((b) => b.classList.remove("active"));
      document.querySelectorAll(".tabpane").forEach((p) => p.classList.remove("active"));
      btn.classList.add("active");
      document.getElementById("tab-" + btn.dataset.tab).classList.add("active");
      localStorage.setItem("gw-tab", btn.dataset.tab);
    };
  });
  const savedTab = localStorage.getItem("gw-tab");
  if (savedTab) {
    const btn = document.querySelector('.tab[data-tab="' + savedTab + '"]');
    if (btn) btn.onclick();
  }

  // ---- editors ----
  const rulesEd = JsoncEditor.create(document.getElementById("rulesEditor"));
  const providersEd = JsoncEditor.create(document.getElementById("providersEditor"));

  function setStatus(id, msg, ok) {
    const el = document.getElementById(id);
    el.textContent = msg;
    el.className = "status " + (ok ? "ok" : "err");
    if (ok) setTimeout(() => (el.textContent = ""), 4000);
  }

  async function loadText(url, editor, statusId) {
    try {
      const r = await fetch(url);
      if (!r.ok) throw new Error(await r.text());
      editor.setValue(await r.text());
      setStatus(statusId, "loaded", true);
    } catch (e) {
      setStatus(statusId, "load failed: " + e.message, false);
    }
  }

  async function saveText(url, editor, statusId) {
    // client-side syntax gate first: mark the offending line inline
    const v = editor.validate();
    if (!v.ok) {
      setStatus(statusId, "not saved — fix the syntax error first", false);
      return;
    }
    try {
      const body = editor.getValue();
      const r = await fetch(url, {
        method: "POST",
        headers: { "Content-Type": "text/plain" },
        body,
      });
      const payload = await r.json().catch(() => ({}));
      if (!r.ok) {
        const msgs = Array.isArray(payload.detail)
          ? payload.detail
              .map((d) => (d.loc ? d.loc.join(".") + ": " : "") + (d.msg || JSON.stringify(d)))
              .join("; ")
          : JSON.stringify(payload.detail || payload);
        editor.markServerError(null, msgs);
        throw new Error(msgs);
      }
      setStatus(statusId, "saved & reloaded", true);
    } catch (e) {
      setStatus(statusId, "save failed: " + e.message, false);
    }
  }

  const RULES_URL = "/v1/config/models-rules";
  const PROVIDERS_URL = "/v1/config/providers";

  document.getElementById("saveRules").onclick = () =>
    saveText(RULES_URL, rulesEd, "rulesStatus");
  document.getElementById("reloadRules").onclick = () =>
    loadText(RULES_URL, rulesEd, "rulesStatus");
  document.getElementById("saveProviders").onclick = () =>
    saveText(PROVIDERS_URL, providersEd, "providersStatus");
  document.getElementById("reloadProviders").onclick = () =>
    loadText(PROVIDERS_URL, providersEd, "providersStatus");

  loadText(RULES_URL, rulesEd, "rulesStatus");
  loadText(PROVIDERS_URL, providersEd, "providersStatus");

  // ---- agent-config exports ----
  function download(name, obj) {
    const blob = new Blob([JSON.stringify(obj, null, 2)], { type: "application/json" });
    const a = document.createElement("a");
    a.href = URL.createObjectURL(blob);
    a.download = name;
    a.click();
    URL.revokeObjectURL(a.href);
    document.getElementById("agentsPreview").textContent = JSON.stringify(obj, null, 2);
  }

  async function exportConfig(kind) {
    const inc = document.getElementById("includeFallback").checked;
    const url =
      "/v1/models/As" + (kind === "opencode" ? "OpenCodeFormat" : "GitHubCopilotFormat") +
      "?includefallback=" + inc;
    const r = await fetch(url);
    const obj = await r.json();
    download(kind === "opencode" ? "opencode-provider.json" : "chatLanguageModels.json", obj);
  }

  document.getElementById("dlOpencode").onclick = () => exportConfig("opencode");
  document.getElementById("dlCopilot").onclick = () => exportConfig("copilot");
})();
