// DOM/WebSocket stubs for the dts_amd visualizer script
function mkEl(id) {
  return {
    id, value: "3", textContent: "", innerHTML: "", className: "",
    style: {}, children: [], childElementCount: 0, lastChild: null,
    disabled: false, files: [],
    addEventListener() {}, appendChild(c) { this.children.push(c); },
    prepend() {}, remove() {}, setAttribute() {}, getAttribute() { return null; },
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
global.setTimeout = (f) => {};  // suppress reconnect timers

const fs = require("fs");
const html = fs.readFileSync(process.argv[2], "utf8");
const script = html.split("<script>")[1].split("</script>")[0];
const assertions = `
function __ev(type, data) { wsInstance.onmessage({ data: JSON.stringify({ type, data }) }); }
wsInstance.onopen();
__ev("search_started", { goal: "g", first_message: "m", total_rounds: 1, config: {} });
__ev("node_added", { id: "root1", parent_id: null, depth: 0, status: "active", strategy: null, user_intent: null, message_count: 1 });
__ev("node_added", { id: "a", parent_id: "root1", depth: 1, status: "active", strategy: "S1", user_intent: null, message_count: 1 });
__ev("node_added", { id: "b", parent_id: "root1", depth: 1, status: "active", strategy: "S2", user_intent: null, message_count: 1 });
__ev("node_updated", { id: "a", status: "active", score: 7.5, individual_scores: [7.5, 7, 8], passed: true });
__ev("nodes_pruned", { ids: ["b"], reasons: { b: "below threshold" } });
__ev("complete", { best_node_id: "a", best_score: 7.5, pruned_count: 1, total_rounds: 1, exploration: { summary: { best_score: 7.5 }, branches: [] } });
if (!(nodes.size === 3)) throw new Error("expected 3 nodes, got " + nodes.size);
if (!(nodes.get("a").score === 7.5)) throw new Error("score not applied");
if (!(nodes.get("b").status === "pruned")) throw new Error("prune not applied");
if (!(nodes.get("a").parent === "root1")) throw new Error("lineage wrong");
console.log("frontend logic OK");
`;
eval(script + assertions);
function unused_ev() {}
