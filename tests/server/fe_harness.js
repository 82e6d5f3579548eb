// DOM/WebSocket stubs for the dts_amd visualizer script: drives every
// WS event type through the real handler code and asserts on the state
// globals (nodes/uiState) the script maintains.
function mkEl(id) {
  const attrs = {};
  return {
    id, value: "3", textContent: "", innerHTML: "", className: "",
    style: {}, children: [], childElementCount: 0, lastChild: null,
    disabled: false, files: [], title: "",
    addEventListener() {}, appendChild(c) { this.children.push(c); },
    prepend() {}, remove() {},
    setAttribute(k, v) { attrs[k] = String(v); },
    getAttribute(k) { return k in attrs ? attrs[k] : null; },
  };
}
const els = {};
global.document = {
  getElementById(id) { return els[id] || (els[id] = mkEl(id)); },
  createElement(t) { return mkEl(t); },
  createElementNS(ns, t) { return mkEl(t); },
};
global.window = global;
global.location = { host: "x", protocol: "http:" };
global.fetch = () => Promise.reject(new Error("offline"));
let wsInstance = null;
global.WebSocket = class {
  constructor(url) { wsInstance = this; this.url = url; }
  send(x) { (this.sent = this.sent || []).push(x); }
};
global.setTimeout = (f) => {};   // suppress reconnect timers
global.setInterval = (f) => 0;   // suppress keepalive/elapsed timers
global.clearInterval = () => {};

const fs = require("fs");
const html = fs.readFileSync(process.argv[2], "utf8");
const script = html.split("<script>")[1].split("</script>")[0];
const assertions = `
function __ev(type, data) { wsInstance.onmessage({ data: JSON.stringify({ type, data }) }); }
function __chk(cond, what) { if (!cond) throw new Error("FAIL: " + what); }
wsInstance.onopen();

// 1. search_started
__ev("search_started", { goal: "g", first_message: "m", total_rounds: 2, config: {} });
__chk(uiState.phase === "starting", "search_started sets phase");

// 2. phase
__ev("phase", { phase: "generating_strategies", message: "gen" });
__chk(uiState.phase === "generating_strategies", "phase applied");

// 3. strategy_generated
__ev("strategy_generated", { index: 1, total: 2, tagline: "S1", description: "d1" });
__ev("strategy_generated", { index: 2, total: 2, tagline: "S2", description: "d2" });
__chk(uiState.strategies.length === 2, "strategies recorded");

// 4. intent_generated
__ev("intent_generated", { label: "skeptic", emotional_tone: "skeptical",
                           cognitive_stance: "challenging", strategy: "S1" });
__chk(uiState.intents.length === 1, "intent recorded");

// 5. round_started
__ev("round_started", { round: 1, total_rounds: 2 });
__chk(uiState.round === 1 && uiState.totalRounds === 2, "round applied");

// 6. node_added (+ tree lineage + layout coordinates)
__ev("node_added", { id: "root1", parent_id: null, depth: 0, status: "active", strategy: null, user_intent: null, message_count: 1 });
__ev("node_added", { id: "a", parent_id: "root1", depth: 1, status: "active", strategy: "S1", user_intent: null, message_count: 3 });
__ev("node_added", { id: "b", parent_id: "root1", depth: 1, status: "active", strategy: "S2", user_intent: "skeptic", message_count: 3 });
__ev("node_added", { id: "a1", parent_id: "a", depth: 2, status: "active", strategy: "S1", user_intent: null, message_count: 5 });
__chk(nodes.size === 4, "4 nodes");
__chk(nodes.get("a").parent === "root1", "lineage");
__chk(typeof nodes.get("a1").x === "number" && typeof nodes.get("a1").y === "number", "layout assigned coordinates");
__chk(nodes.get("a1").y > nodes.get("a").y, "depth ordering in layout");
__chk(nodes.get("a").x !== nodes.get("b").x, "siblings separated");

// 7. node_updated
__ev("node_updated", { id: "a", status: "active", score: 7.5, individual_scores: [7.5, 7, 8], passed: true });
__chk(nodes.get("a").score === 7.5, "score applied");
__chk(uiState.scored === 1, "scored counter");
__chk(uiState.bestScore === 7.5, "best score tracked");

// 8. nodes_pruned (with reasons)
__ev("nodes_pruned", { ids: ["b"], reasons: { b: "below threshold 6.5" } });
__chk(nodes.get("b").status === "pruned", "prune applied");
__chk(nodes.get("b").pruneReason.indexOf("threshold") >= 0, "prune reason kept");
__chk(uiState.pruned === 1, "pruned counter");

// 9. token_update
__ev("token_update", { totals: { input_tokens: 1200, output_tokens: 340, total_cost_usd: 0.0 } });
__chk(uiState.tokens.input === 1200 && uiState.tokens.output === 340, "token totals");

// 10. research_log
__ev("research_log", { message: "searching sources" });
__chk(uiState.researchLines.length === 1, "research line kept");

// 11. pong
__ev("pong", {});
__chk(uiState.pongs === 1, "pong counted");

// 12. complete (selects best node, exploration drives details)
__ev("complete", { best_node_id: "a", best_score: 7.5, pruned_count: 1,
  total_rounds: 2,
  exploration: { summary: { best_score: 7.5 }, branches: [
    { id: "a", parent_id: null, status: "active", depth: 1,
      strategy: { tagline: "S1" },
      scores: { aggregated: 7.5, individual: [7.5, 7, 8],
                critiques: { weaknesses: ["w1"], strengths: ["s1"], key_moment: "km" } },
      trajectory: [ { role: "user", content: "hi" }, { role: "assistant", content: "yo" } ] },
  ] } });
__chk(uiState.phase === "complete", "complete phase");
__chk(uiState.selected === "a", "best node selected");

// details panel renders the trajectory + judge tabs without throwing
uiState.tab = "judge"; renderTab();
uiState.tab = "conv"; renderTab();
uiState.tab = "usage"; renderTab();
__chk(document.getElementById("tabbody").innerHTML.indexOf("1,200") >= 0
      || document.getElementById("tabbody").innerHTML.indexOf("1200") >= 0,
      "usage panel shows tokens");

// 13. error resets start button
__ev("error", { message: "boom" });
__chk(uiState.errors.length === 1, "error recorded");

// start button sends a protocol-correct start_search
document.getElementById("f-goal").value = "G2";
// (button listeners are wired but stub addEventListener drops them;
// exercise requestConfig directly — it is the wire contract)
const cfg = requestConfig();
__chk(cfg.goal === "G2" && typeof cfg.user_variability === "boolean"
      && typeof cfg.reasoning_enabled === "boolean", "request config shape");

// pan/zoom state math
view.k = 1; view.x = 0; view.y = 0; fitView();
__chk(view.k > 0 && isFinite(view.x) && isFinite(view.y), "fitView finite");

console.log("frontend logic OK");
`;
eval(script + assertions);
function unused_ev() {}
