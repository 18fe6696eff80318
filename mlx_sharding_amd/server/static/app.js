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
function mdLite(text) {
  // minimal markdown: code blocks, inline code, bold, newlines
  const esc = text.replace(/&/g, "&amp;").replace(/</g, "&lt;").replace(/>/g, "&gt;");
  return esc
    .replace(/```([\s\S]*?)```/g, "<pre><code>$1</code></pre>")
    .replace(/`([^`]+)`/g, "<code>$1</code>")
    .replace(/\*\*([^*]+)\*\*/g, "<strong>$1</strong>")
    .replace(/\n/g, "<br>");
}
function renderMessages() {
  const div = $("messages");
  div.innerHTML = "";
  if (!current) return;
  for (const m of current.messages) {
    const el = document.createElement("div");
    el.className = `msg ${m.role}`;
    el.innerHTML = `<div class="role">${m.role}</div><div class="content">${mdLite(m.content)}</div>`;
    div.appendChild(el);
  }
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
