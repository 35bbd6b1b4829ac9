// Config editor logic (load/save via /v1/config/*, agent-config exports).
// Functional parity with the reference's editor.js:169-334, rebuilt without
// CDN CodeMirror (plain textarea; server-side validation is authoritative).

(function () {
  "use strict";

  // ---- theme ----
  const root = document.documentElement;
  function applyTheme(t) {
    root.dataset.theme = t;
    localStorage.setItem("gw-theme", t);
  }
  applyTheme(localStorage.getItem("gw-theme") || "dark");
  document.getElementById("themeToggle").onclick = () =>
    applyTheme(root.dataset.theme === "dark" ? "light" : "dark");

  // ---- tabs ----
  document.querySelectorAll(".tab").forEach((btn) => {
    btn.onclick = () => {
      document.querySelectorAll(".tab").forEach((b) => b.classList.remove("active"));
      document.querySelectorAll(".tabpane").forEach((p) => p.classList.remove("active"));
      btn.classList.add("active");
      document.getElementById("tab-" + btn.dataset.tab).classList.add("active");
    };
  });

  function setStatus(id, msg, ok) {
    const el = document.getElementById(id);
    el.textContent = msg;
    el.className = "status " + (ok ? "ok" : "err");
    if (ok) setTimeout(() => (el.textContent = ""), 4000);
  }

  async function loadText(url, areaId, statusId) {
    try {
      const r = await fetch(url);
      if (!r.ok) throw new Error(await r.text());
      document.getElementById(areaId).value = await r.text();
      setStatus(statusId, "loaded", true);
    } catch (e) {
      setStatus(statusId, "load failed: " + e.message, false);
    }
  }

  async function saveText(url, areaId, statusId) {
    try {
      const body = document.getElementById(areaId).value;
      const r = await fetch(url, {
        method: "POST",
        headers: { "Content-Type": "text/plain" },
        body,
      });
      const payload = await r.json().catch(() => ({}));
      if (!r.ok) {
        const msgs = Array.isArray(payload.detail)
          ? payload.detail.map((d) => d.msg || JSON.stringify(d)).join("; ")
          : JSON.stringify(payload.detail || payload);
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
    saveText(RULES_URL, "rulesText", "rulesStatus");
  document.getElementById("reloadRules").onclick = () =>
    loadText(RULES_URL, "rulesText", "rulesStatus");
  document.getElementById("saveProviders").onclick = () =>
    saveText(PROVIDERS_URL, "providersText", "providersStatus");
  document.getElementById("reloadProviders").onclick = () =>
    loadText(PROVIDERS_URL, "providersText", "providersStatus");

  loadText(RULES_URL, "rulesText", "rulesStatus");
  loadText(PROVIDERS_URL, "providersText", "providersStatus");

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
