// Usage-stats dashboard logic. Parity with the reference usage-stats.js:
// per-period, per-model aggregate tables (incl. derived Cost/Mtok), raw
// records pagination (25/page), plus the MI355X engine-stats tab.

(function () {
  "use strict";

  const root = document.documentElement;
  const themeSel = document.getElementById("themeSelector");
  function applyTheme(t) {
    root.dataset.theme = t;
    localStorage.setItem("gw-theme", t);
    if (themeSel.value !== t) themeSel.value = t;
  }
  applyTheme(localStorage.getItem("gw-theme") || "dark");
  themeSel.addEventListener("change", () => applyTheme(themeSel.value));

  document.querySelectorAll(".tab").forEach((btn) => {
    btn.onclick = () => {
      document.querySelectorAll(".tab").forEach((b) => b.classList.remove("active"));
      document.querySelectorAll(".tabpane").forEach((p) => p.classList.remove("active"));
      btn.classList.add("active");
      document.getElementById("tab-" + btn.dataset.tab).classList.add("active");
    };
  });

  const fmt = (n) => (n == null ? "—" : Number(n).toLocaleString());
  const fmtCost = (n) => (n == null || n === 0 ? "—" : "$" + Number(n).toFixed(4));

  function table(headers, rows) {
    let h = "<table><thead><tr>" + headers.map((x) => `<th>${x}</th>`).join("") +
      "</tr></thead><tbody>";
    for (const r of rows) h += "<tr>" + r.map((x) => `<td>${x}</td>`).join("") + "</tr>";
    return h + "</tbody></table>";
  }

  // ---- statistics ----
  async function loadStats() {
    const period = document.getElementById("period").value;
    const r = await fetch(`/v1/api/usage-stats/${period}`);
    const data = await r.json();
    const byPeriod = {};
    for (const row of data) {
      (byPeriod[row.time_period] = byPeriod[row.time_period] || []).push(row);
    }
    const el = document.getElementById("statsTables");
    el.innerHTML = "";
    const periods = Object.keys(byPeriod).sort().reverse();
    if (!periods.length) { el.innerHTML = "<p class='hint'>No usage recorded yet.</p>"; return; }
    for (const p of periods) {
      const rows = byPeriod[p].map((r) => {
        const costPerM = r.total_tokens > 0 && r.cost > 0
          ? "$" + ((r.cost / r.total_tokens) * 1e6).toFixed(3) : "—";
        return [r.model || "—", fmt(r.count), fmt(r.prompt_tokens), fmt(r.completion_tokens),
                fmt(r.reasoning_tokens), fmt(r.cached_tokens), fmt(r.total_tokens),
                fmtCost(r.cost), costPerM];
      });
      el.innerHTML += `<h3>${p}</h3>` + table(
        ["Model", "Reqs", "Prompt", "Completion", "Reasoning", "Cached", "Total", "Cost", "Cost/Mtok"],
        rows);
    }
  }
  document.getElementById("refreshStats").onclick = loadStats;
  document.getElementById("period").onchange = loadStats;

  // ---- records ----
  const PAGE = 25;
  let offset = 0;
  async function loadRecords() {
    const r = await fetch(`/v1/api/usage-records?limit=${PAGE}&offset=${offset}`);
    const { records, total } = await r.json();
    document.getElementById("pageInfo").textContent =
      total ? `${offset + 1}–${Math.min(offset + PAGE, total)} of ${total}` : "no records";
    document.getElementById("recordsTable").innerHTML = table(
      ["Time", "Model", "Provider", "Prompt", "Completion", "Reasoning", "Cached", "Total", "Cost"],
      records.map((r) => [
        (r.timestamp || "").replace("T", " ").slice(0, 19), r.model || "—", r.provider || "—",
        fmt(r.prompt_tokens), fmt(r.completion_tokens), fmt(r.reasoning_tokens),
        fmt(r.cached_tokens), fmt(r.total_tokens), fmtCost(r.cost)]));
  }
  document.getElementById("prevPage").onclick = () => {
    offset = Math.max(0, offset - PAGE); loadRecords();
  };
  document.getElementById("nextPage").onclick = () => { offset += PAGE; loadRecords(); };

  // ---- engines ----
  async function loadEngines() {
    const r = await fetch("/v1/api/engine-stats");
    const { engines } = await r.json();
    document.getElementById("enginesTable").innerHTML = engines.length
      ? table(
          ["Engine", "Device", "Model", "Waiting", "Prefilling", "Running",
           "KV free/total", "Requests", "Finished", "Failed",
           "Prefill toks", "Decode toks", "Prefix hits", "Prefix toks"],
          engines.map((e) => [
            e.engine, e.device, e.model, e.waiting ?? "-", e.prefilling ?? "-",
            e.running ?? "-",
            `${e.kv_blocks_free ?? "-"}/${e.kv_blocks_total ?? "-"}`, fmt(e.requests),
            fmt(e.finished), fmt(e.failed), fmt(e.prefill_tokens), fmt(e.decode_tokens),
            fmt(e.prefix_cache_hits ?? 0), fmt(e.prefix_cached_tokens ?? 0)]))
      : "<p class='hint'>No local engines running.</p>";
  }
  document.getElementById("refreshEngines").onclick = loadEngines;

  loadStats();
  loadRecords();
  loadEngines();
})();
