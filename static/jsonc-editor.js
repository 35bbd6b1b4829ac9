// Self-hosted JSON-with-comments code editor (no CDN): a transparent
// <textarea> over a synced highlight layer, with a line-number gutter,
// live JSONC validation (inline error marker at the offending line) and
// CSS-variable themes. Functional stand-in for the reference's CodeMirror 5
// setup (/root/reference/static/editor.js:1-361) built from scratch.
//
// Dialect matches the server (llmapigateway_amd/config/jsonc.py): // and
// /* */ comments, trailing commas, otherwise strict JSON.

(function () {
  "use strict";

  // ---- JSONC tokenizer (for highlighting) ----
  const TOKEN_RE = new RegExp(
    [
      "(\\/\\/[^\\n]*|\\/\\*[\\s\\S]*?(?:\\*\\/|$))", // 1 comment
      '("(?:[^"\\\\\\n]|\\\\.)*"?)',                   // 2 string
      "(-?\\b\\d+(?:\\.\\d+)?(?:[eE][+-]?\\d+)?\\b)", // 3 number
      "\\b(true|false|null)\\b",                        // 4 keyword
      "([{}\\[\\],:])",                                  // 5 punctuation
    ].join("|"),
    "g"
  );

  function esc(s) {
    return s.replace(/&/g, "&amp;").replace(/</g, "&lt;").replace(/>/g, "&gt;");
  }

  function highlight(src) {
    let out = "";
    let last = 0;
    let m;
    TOKEN_RE.lastIndex = 0;
    while ((m = TOKEN_RE.exec(src)) !== null) {
      out += esc(src.slice(last, m.index));
      const [tok, com, str, num, kw, punct] = m;
      let cls = punct ? "p" : kw ? "k" : num ? "n" : com ? "c" : "s";
      if (str) {
        // a string directly followed by ':' is a property key
        const rest = src.slice(m.index + tok.length);
        if (/^\s*:/.test(rest)) cls = "key";
      }
      out += '<span class="tk-' + cls + '">' + esc(tok) + "</span>";
      last = m.index + tok.length;
    }
    out += esc(src.slice(last));
    return out;
  }

  // ---- JSONC validation (mirrors config/jsonc.py: strip comments and
  // trailing commas offset-preserving, then strict JSON.parse) ----
  function stripJsonc(text) {
    const out = text.split("");
    let i = 0;
    const n = text.length;
    let inStr = false;
    while (i < n) {
      const c = text[i];
      if (inStr) {
        if (c === "\\") i += 1;
        else if (c === '"') inStr = false;
        i += 1;
        continue;
      }
      if (c === '"') {
        inStr = true;
        i += 1;
        continue;
      }
      if (c === "/" && text[i + 1] === "/") {
        while (i < n && text[i] !== "\n") out[i++] = " ";
        continue;
      }
      if (c === "/" && text[i + 1] === "*") {
        out[i] = out[i + 1] = " ";
        i += 2;
        while (i < n && !(text[i] === "*" && text[i + 1] === "/")) {
          if (text[i] !== "\n") out[i] = " ";
          i += 1;
        }
        if (i < n) { out[i] = out[i + 1] = " "; i += 2; }
        continue;
      }
      if (c === ",") {
        // trailing comma: next non-space/comment char is } or ]
        let j = i + 1;
        while (j < n) {
          const d = text[j];
          if (/\s/.test(d)) { j += 1; continue; }
          if (d === "/" && text[j + 1] === "/") { while (j < n && text[j] !== "\n") j += 1; continue; }
          if (d === "/" && text[j + 1] === "*") { j += 2; while (j < n && !(text[j] === "*" && text[j + 1] === "/")) j += 1; j += 2; continue; }
          break;
        }
        if (j < n && (text[j] === "}" || text[j] === "]")) out[i] = " ";
        i += 1;
        continue;
      }
      i += 1;
    }
    return out.join("");
  }

  function validateJsonc(text) {
    if (!text.trim()) return { ok: true };
    const stripped = stripJsonc(text);
    try {
      JSON.parse(stripped);
      return { ok: true };
    } catch (e) {
      // "... at position 123 (line 4 column 5)" or "... at position 123"
      let line = null;
      let msg = e.message;
      let m = msg.match(/line (\d+)/);
      if (m) line = parseInt(m[1], 10);
      else if ((m = msg.match(/position (\d+)/)))
        line = stripped.slice(0, parseInt(m[1], 10)).split("\n").length;
      return { ok: false, line, message: msg };
    }
  }

  // ---- the editor component ----
  function create(host, opts) {
    opts = opts || {};
    host.classList.add("jsonc-editor");
    host.innerHTML =
      '<div class="je-gutter"></div>' +
      '<div class="je-body">' +
      '<pre class="je-hl" aria-hidden="true"><code></code></pre>' +
      '<textarea class="je-input" spellcheck="false" autocapitalize="off" autocomplete="off"></textarea>' +
      "</div>" +
      '<div class="je-footer"><span class="je-pos"></span><span class="je-err"></span></div>';
    const gutter = host.querySelector(".je-gutter");
    const hl = host.querySelector(".je-hl");
    const code = host.querySelector(".je-hl code");
    const input = host.querySelector(".je-input");
    const posEl = host.querySelector(".je-pos");
    const errEl = host.querySelector(".je-err");
    let errLine = null;
    let timer = null;

    function renderGutter() {
      const lines = input.value.split("\n").length;
      let g = "";
      for (let i = 1; i <= lines; i += 1)
        g += '<div class="je-ln' + (i === errLine ? " je-ln-err" : "") + '">' + i + "</div>";
      gutter.innerHTML = g;
    }

    function render() {
      code.innerHTML = highlight(input.value) + "\n"; // trailing nl keeps heights equal
      renderGutter();
    }

    function validate() {
      const v = validateJsonc(input.value);
      errLine = v.ok ? null : v.line;
      errEl.textContent = v.ok ? "" : "✗ " + v.message;
      host.classList.toggle("je-invalid", !v.ok);
      renderGutter();
      if (opts.onValidate) opts.onValidate(v);
      return v;
    }

    function caretPos() {
      const upto = input.value.slice(0, input.selectionStart).split("\n");
      posEl.textContent = "Ln " + upto.length + ", Col " + (upto[upto.length - 1].length + 1);
    }

    input.addEventListener("input", () => {
      render();
      caretPos();
      clearTimeout(timer);
      timer = setTimeout(validate, 300);
    });
    input.addEventListener("scroll", () => {
      hl.scrollTop = input.scrollTop;
      hl.scrollLeft = input.scrollLeft;
      gutter.scrollTop = input.scrollTop;
    });
    ["keyup", "click"].forEach((ev) => input.addEventListener(ev, caretPos));
    // soft-tab insertion (editor nicety CodeMirror provided)
    input.addEventListener("keydown", (e) => {
      if (e.key === "Tab") {
        e.preventDefault();
        const s = input.selectionStart;
        input.setRangeText("  ", s, input.selectionEnd, "end");
        input.dispatchEvent(new Event("input"));
      }
    });

    render();
    return {
      getValue: () => input.value,
      setValue: (v) => {
        input.value = v;
        render();
        validate();
      },
      validate,
      markServerError: (line, message) => {
        errLine = line || null;
        errEl.textContent = "✗ " + message;
        host.classList.add("je-invalid");
        renderGutter();
      },
      textarea: input,
    };
  }

  const api = { create, validateJsonc, highlight, stripJsonc };
  if (typeof window !== "undefined") window.JsoncEditor = api;
  if (typeof module !== "undefined" && module.exports) module.exports = api;
})();
