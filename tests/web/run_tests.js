#!/usr/bin/env node
/* Zero-dependency JS test runner (node >= 12; the image has no vitest).
 * Mirrors the reference's web/tests/ coverage areas: URL-building parity,
 * the interceptor's submission decision, DistributedValue widget logic,
 * divider socket morphing. Run: node tests/web/run_tests.js */
"use strict";
const assert = require("assert");
const path = require("path");

const D = require(path.join(__dirname, "..", "..", "comfyui_distributed_amd",
                            "server", "static", "js", "distributed.js"));

const tests = [];
function test(name, fn) { tests.push([name, fn]); }

// ---- urlUtils parity (reference web/tests/urlUtils.test.js) --------------
test("buildWorkerUrl basic", () => {
  assert.strictEqual(D.buildWorkerUrl({ host: "10.0.0.2", port: 8189 }),
                     "http://10.0.0.2:8189");
});
test("buildWorkerUrl blank host -> localhost", () => {
  assert.strictEqual(D.buildWorkerUrl({ host: "", port: 8189 }),
                     "http://localhost:8189");
});
test("buildWorkerUrl strips scheme and trailing slash", () => {
  assert.strictEqual(D.buildWorkerUrl({ host: "http://box/", port: 9000 }),
                     "http://box:9000");
});
test("buildWorkerUrl embedded port wins", () => {
  assert.strictEqual(D.buildWorkerUrl({ host: "1.2.3.4:9000", port: 8189 }),
                     "http://1.2.3.4:9000");
});
test("buildWorkerUrl cloud host -> https, no port", () => {
  assert.strictEqual(
    D.buildWorkerUrl({ host: "abc.trycloudflare.com", port: 8189 }),
    "https://abc.trycloudflare.com");
  assert.strictEqual(
    D.buildWorkerUrl({ host: "x-8189.proxy.runpod.net", port: 8189 }),
    "https://x-8189.proxy.runpod.net");
});
test("splitHostPort ipv6", () => {
  assert.deepStrictEqual(D.splitHostPort("[::1]:8189"), ["::1", 8189]);
  assert.deepStrictEqual(D.splitHostPort("[::1]"), ["::1", null]);
});
test("buildMasterUrl defaults + cloud", () => {
  assert.strictEqual(D.buildMasterUrl({}, 8188), "http://127.0.0.1:8188");
  assert.strictEqual(D.buildMasterUrl({ host: "t.trycloudflare.com" }),
                     "https://t.trycloudflare.com");
});

// ---- interceptor decision (reference web/executionUtils.js:6-151) --------
const collectorPrompt = {
  "1": { class_type: "LoadImage", inputs: { image: "x.png" } },
  "2": { class_type: "DistributedCollector",
         inputs: { images: ["1", 0] } },
};
test("no distributed nodes -> vanilla local queue", () => {
  const d = D.decideSubmission({
    prompt: { "1": { class_type: "KSampler", inputs: {} } },
    enabledWorkers: [{ id: "w1" }], probeResults: { w1: true },
  });
  assert.strictEqual(d.mode, "local");
});
test("all workers offline -> master-only fallback", () => {
  const d = D.decideSubmission({
    prompt: collectorPrompt,
    enabledWorkers: [{ id: "w1" }, { id: "w2" }],
    probeResults: { w1: false, w2: false },
  });
  assert.strictEqual(d.mode, "local");
  assert.ok(/offline/.test(d.reason));
});
test("subset online -> distributed with active ids only", () => {
  const d = D.decideSubmission({
    prompt: collectorPrompt,
    enabledWorkers: [{ id: "w1" }, { id: "w2" }],
    probeResults: { w1: true, w2: false },
  });
  assert.strictEqual(d.mode, "distributed");
  assert.deepStrictEqual(d.activeIds, ["w1"]);
});
test("dead cloudflare master blocks execution", () => {
  const d = D.decideSubmission({
    prompt: collectorPrompt,
    enabledWorkers: [{ id: "w1" }], probeResults: { w1: true },
    masterHost: "abc.trycloudflare.com", masterReachable: false,
  });
  assert.strictEqual(d.mode, "blocked");
});
test("live cloudflare master proceeds", () => {
  const d = D.decideSubmission({
    prompt: collectorPrompt,
    enabledWorkers: [{ id: "w1" }], probeResults: { w1: true },
    masterHost: "abc.trycloudflare.com", masterReachable: true,
  });
  assert.strictEqual(d.mode, "distributed");
});
test("non-cloudflare master ignores reachability", () => {
  const d = D.decideSubmission({
    prompt: collectorPrompt,
    enabledWorkers: [{ id: "w1" }], probeResults: { w1: true },
    masterHost: "10.0.0.5", masterReachable: false,
  });
  assert.strictEqual(d.mode, "distributed");
});

// ---- DistributedValue widgets (reference web/distributedValue.js) --------
const valuePrompt = () => ({
  "7": { class_type: "DistributedValue",
         inputs: { default_value: "20",
                   worker_values: JSON.stringify({ _type: "INT", "1": 30 }) } },
  "8": { class_type: "KSampler", inputs: { steps: ["7", 0] } },
});
test("valueWidgetModel: one row per enabled worker, 1-indexed", () => {
  const rows = D.valueWidgetModel(valuePrompt(), [
    { id: "a", name: "GPU 1" }, { id: "b" }]);
  assert.strictEqual(rows.length, 1);
  const r = rows[0];
  assert.strictEqual(r.valueType, "INT");
  assert.strictEqual(r.defaultValue, "20");
  assert.deepStrictEqual(r.workers.map(w => w.key), ["1", "2"]);
  assert.strictEqual(r.workers[0].value, "30");  // existing override
  assert.strictEqual(r.workers[1].value, "");    // unset
});
test("applyValueEdits: typed coercion + empty clears override", () => {
  const p = valuePrompt();
  D.applyValueEdits(p, "7", "25", { "1": "", "2": "40.7" });
  const store = JSON.parse(p["7"].inputs.worker_values);
  assert.strictEqual(p["7"].inputs.default_value, "25");
  assert.strictEqual(store["1"], undefined);  // cleared
  assert.strictEqual(store["2"], 40);         // INT coercion truncates
  assert.strictEqual(store._type, "INT");
});
test("coerceValue matches the python node", () => {
  assert.strictEqual(D.coerceValue("3.9", "INT"), 3);
  assert.strictEqual(D.coerceValue("3.9", "FLOAT"), 3.9);
  assert.strictEqual(D.coerceValue("3.9", "STRING"), "3.9");
  assert.strictEqual(D.coerceValue("abc", "INT"), "abc");  // unparsable kept
});

// ---- divider socket morphing (reference web/image_batch_divider.js) ------
test("dividerOutputs follows divide_by, clamped 1..10", () => {
  assert.deepStrictEqual(
    D.dividerOutputs("ImageBatchDivider", 3).map(o => o.name),
    ["batch_1", "batch_2", "batch_3"]);
  assert.strictEqual(D.dividerOutputs("ImageBatchDivider", 99).length, 10);
  assert.strictEqual(D.dividerOutputs("ImageBatchDivider", 0).length, 1);
  assert.strictEqual(D.dividerOutputs("AudioBatchDivider", 2)[0].name,
                     "audio_1");
  assert.strictEqual(D.dividerOutputs("KSampler", 2), null);
});
test("danglingDividerLinks flags consumers past divide_by", () => {
  const p = {
    "1": { class_type: "ImageBatchDivider",
           inputs: { images: ["0", 0], divide_by: 2 } },
    "2": { class_type: "PreviewImage", inputs: { images: ["1", 1] } },
    "3": { class_type: "PreviewImage", inputs: { images: ["1", 2] } },
  };
  const bad = D.danglingDividerLinks(p);
  assert.strictEqual(bad.length, 1);
  assert.strictEqual(bad[0].node, "3");
  assert.strictEqual(bad[0].available, 2);
});

// ---- runner --------------------------------------------------------------
let fails = 0;
for (const [name, fn] of tests) {
  try {
    fn();
    console.log("ok   " + name);
  } catch (e) {
    fails++;
    console.error("FAIL " + name + "\n     " + e.message);
  }
}
console.log(`\n${tests.length - fails}/${tests.length} passed`);
process.exit(fails ? 1 : 0);
