/* Chat client for the OpenAI-compatible API: SSE streaming, localStorage
 * sessions + settings, regenerate.  Capability parity with the reference
 * web UI (/root/reference/shard/static/app.js): settings persistence,
 * session list, streaming consumption, stop, regenerate. */

const $ = (id) => document.getElementById(id);

const SETTING_IDS = ["endpoint", "api-key", "model", "max-tokens",
  "temperature", "top-p", "stop", "system-prompt"];

function loadSettings() {
  const saved = JSON.parse(localStorage.getItem("mlxs_settings") || "{}");
  for (const id of SETTING_IDS) {
    if (saved[id] !== undefined) $(id).value = saved[id];
    $(id).addEventListener("change", saveSettings);
  }
}
function saveSettings() {
  const out = {};
  for (const id of SETTING_IDS) out[id] = $(id).value;
  localStorage.setItem("mlxs_settings", JSON.stringify(out));
}

// ---- sessions --------------------------------------------------------------
let sessions = JSON.parse(localStorage.getItem("mlxs_sessions") || "[]");
let current = null;   // {id, title, messages: [{role, content}]}
let aborter = null;

function persist() {
  localStorage.setItem("mlxs_sessions", JSON.stringify(sessions));
}
function newSession() {
  current = { id: Date.now().toString(36), title: "New chat", messages: [] };
  sessions.unshift(current);
  persist(); renderSessions(); renderMessages();
}
function renderSessions() {
  const ul = $("sessions");
  ul.innerHTML = "";
  for (const s of sessions) {
    const li = document.createElement("li");
    li.textContent = s.title;
    li.className = current && s.id === current.id ? "active" : "";
    li.onclick = () => { current = s; renderSessions(); renderMessages(); };
    const del = document.createElement("span");
    del.textContent = "✕";
    del.className = "del";
    del.onclick = (e) => {
      e.stopPropagation();
      sessions = sessions.filter((x) => x.id !== s.id);
      if (current && current.id === s.id) current = sessions[0] || null;
      persist(); renderSessions(); renderMessages();
    };
    li.appendChild(del);
    ul.appendChild(li);
  }
}
function escapeHtml(s) {
  return s.replace(/&/g, "&amp;").replace(/</g, "&lt;").replace(/>/g, "&gt;");
}
function mdInline(s) {
  // inline code, links, bold, italic (input already HTML-escaped)
  return s
    .replace(/`([^`]+)`/g, "<code>$1</code>")
    .replace(/\[([^\]]+)\]\((https?:[^)\s]+)\)/g,
      '<a href="$2" target="_blank" rel="noopener">$1</a>')
    .replace(/\*\*([^*]+)\*\*/g, "<strong>$1</strong>")
    .replace(/(^|[^*])\*([^*\s][^*]*)\*/g, "$1<em>$2</em>");
}
function mdRender(text) {
  // hand-written markdown: fenced code, headings, lists, quotes, hr
  const out = [];
  const parts = text.split(/```/);
  for (let i = 0; i < parts.length; i++) {
    if (i % 2 === 1) {  // inside a code fence (first line = language tag)
      const body = parts[i].replace(/^[^\n]*\n?/, "");
      out.push(`<pre><code>${escapeHtml(body)}</code></pre>`);
      continue;
    }
    const lines = escapeHtml(parts[i]).split("\n");
    let list = null;  // "ul" | "ol" while inside a list
    const closeList = () => { if (list) { out.push(`</${list}>`); list = null; } };
    for (const line of lines) {
      let m;
      if ((m = line.match(/^(#{1,6})\s+(.*)$/))) {
        closeList();
        const h = Math.min(m[1].length + 2, 6);  // h3..h6 inside chat
        out.push(`<h${h}>${mdInline(m[2])}</h${h}>`);
      } else if ((m = line.match(/^\s*[-*]\s+(.*)$/))) {
        if (list !== "ul") { closeList(); out.push("<ul>"); list = "ul"; }
        out.push(`<li>${mdInline(m[1])}</li>`);
      } else if ((m = line.match(/^\s*\d+[.)]\s+(.*)$/))) {
        if (list !== "ol") { closeList(); out.push("<ol>"); list = "ol"; }
        out.push(`<li>${mdInline(m[1])}</li>`);
      } else if (/^\s*(---|\*\*\*)\s*$/.test(line)) {
        closeList(); out.push("<hr>");
      } else if ((m = line.match(/^&gt;\s?(.*)$/))) {
        closeList(); out.push(`<blockquote>${mdInline(m[1])}</blockquote>`);
      } else if (line.trim() === "") {
        closeList();
      } else {
        closeList(); out.push(`<p>${mdInline(line)}</p>`);
      }
    }
    closeList();
  }
  return out.join("\n");
}

let editing = null;  // index of the user message being edited

function renderMessages() {
  const div = $("messages");
  div.innerHTML = "";
  if (!current) return;
  current.messages.forEach((m, idx) => {
    const el = document.createElement("div");
    el.className = `msg ${m.role}`;
    if (editing === idx && m.role === "user") {
      // edit-resend (reference app.js:102-140): edit the prompt in
      // place, drop everything after it, regenerate
      el.innerHTML = `<div class="role">${m.role}</div>`;
      const ta = document.createElement("textarea");
      ta.className = "edit-area";
      ta.value = m.content;
      const save = document.createElement("button");
      save.textContent = "Save & resend";
      save.onclick = () => {
        m.content = ta.value.trim();
        current.messages = current.messages.slice(0, idx + 1);
        editing = null;
        persist(); renderMessages();
        generate();
      };
      const cancel = document.createElement("button");
      cancel.textContent = "Cancel";
      cancel.onclick = () => { editing = null; renderMessages(); };
      const row = document.createElement("div");
      row.className = "edit-actions";
      row.append(save, cancel);
      el.append(ta, row);
    } else {
      el.innerHTML = `<div class="role">${m.role}</div>` +
        `<div class="content">${mdRender(m.content)}</div>`;
      if (m.role === "user" && !aborter) {
        const edit = document.createElement("button");
        edit.textContent = "edit";
        edit.className = "msg-edit";
        edit.onclick = () => { editing = idx; renderMessages(); };
        el.appendChild(edit);
      }
    }
    div.appendChild(el);
  });
  div.scrollTop = div.scrollHeight;
}

// ---- generation ------------------------------------------------------------
async function generate() {
  if (!current || aborter) return;
  const sys = $("system-prompt").value.trim();
  const messages = [];
  if (sys) messages.push({ role: "system", content: sys });
  messages.push(...current.messages);

  const stops = $("stop").value.split(",").map((s) => s.trim()).filter(Boolean);
  const body = {
    model: $("model").value || "default_model",
    messages,
    stream: true,
    max_tokens: parseInt($("max-tokens").value) || 512,
    temperature: parseFloat($("temperature").value) || 0,
    top_p: parseFloat($("top-p").value) || 1,
  };
  if (stops.length) body.stop = stops;

  const headers = { "Content-Type": "application/json" };
  const key = $("api-key").value;
  if (key) headers["Authorization"] = `Bearer ${key}`;

  current.messages.push({ role: "assistant", content: "" });
  const msg = current.messages[current.messages.length - 1];
  renderMessages();
  aborter = new AbortController();
  $("stop-gen").disabled = false;
  try {
    const resp = await fetch($("endpoint").value, {
      method: "POST", headers, body: JSON.stringify(body),
      signal: aborter.signal,
    });
    const reader = resp.body.getReader();
    const decoder = new TextDecoder();
    let buf = "";
    for (;;) {
      const { done, value } = await reader.read();
      if (done) break;
      buf += decoder.decode(value, { stream: true });
      const lines = buf.split("\n\n");
      buf = lines.pop();
      for (const line of lines) {
        if (!line.startsWith("data: ")) continue;
        const payload = line.slice(6);
        if (payload === "[DONE]") continue;
        try {
          const chunk = JSON.parse(payload);
          const delta = chunk.choices?.[0]?.delta?.content
            ?? chunk.choices?.[0]?.text ?? "";
          if (delta) { msg.content += delta; renderMessages(); }
        } catch { /* partial frame */ }
      }
    }
  } catch (e) {
    if (e.name !== "AbortError") msg.content += `\n[error: ${e.message}]`;
  } finally {
    aborter = null;
    $("stop-gen").disabled = true;
    if (current.title === "New chat" && current.messages.length >= 1) {
      const first = current.messages.find((m) => m.role === "user");
      if (first) current.title = first.content.slice(0, 40);
    }
    persist(); renderSessions(); renderMessages();
  }
}

// ---- wiring ----------------------------------------------------------------
$("composer").addEventListener("submit", (e) => {
  e.preventDefault();
  const text = $("input").value.trim();
  if (!text) return;
  if (!current) newSession();
  current.messages.push({ role: "user", content: text });
  $("input").value = "";
  persist(); renderMessages();
  generate();
});
$("input").addEventListener("keydown", (e) => {
  if (e.key === "Enter" && !e.shiftKey) {
    e.preventDefault();
    $("composer").requestSubmit();
  }
});
$("new-chat").onclick = newSession;
$("stop-gen").onclick = () => aborter && aborter.abort();
$("regen").onclick = () => {
  if (!current) return;
  while (current.messages.length &&
         current.messages[current.messages.length - 1].role === "assistant") {
    current.messages.pop();
  }
  persist(); renderMessages();
  generate();
};

loadSettings();
if (sessions.length) current = sessions[0];
renderSessions();
renderMessages();
