/* Shared browser/node logic for the distributed panel.
 *
 * MI355X-native counterpart of the reference's editor-embedded JS
 * (web/urlUtils.js, web/executionUtils.js:6-151, web/distributedValue.js,
 * web/image_batch_divider.js — behavior parity, different host: this
 * framework's panel drives prompt JSON directly instead of LiteGraph
 * nodes). UMD so the same file runs as a <script> in panel.html and under
 * plain `node` in tests/web/ (node 12, no bundler, no deps).
 */
(function (root, factory) {
  if (typeof module === "object" && module.exports) module.exports = factory();
  else root.Distributed = factory();
})(typeof self !== "undefined" ? self : this, function () {
  "use strict";

  var NODE_CLASSES = {
    COLLECTOR: "DistributedCollector",
    UPSCALE: "UltimateSDUpscaleDistributed",
    VALUE: "DistributedValue",
    SEED: "DistributedSeed",
    MODEL_NAME: "DistributedModelName",
  };

  // ---- url building (parity with server/network.py:41-105) --------------
  function normalizeHost(host) {
    host = String(host || "").trim();
    ["http://", "https://"].forEach(function (p) {
      if (host.indexOf(p) === 0) host = host.slice(p.length);
    });
    return host.replace(/\/+$/, "");
  }

  function splitHostPort(hostport) {
    var h = normalizeHost(hostport);
    if (h[0] === "[") {
      var idx = h.indexOf("]");
      var host = h.slice(1, idx);
      var rest = h.slice(idx + 1);
      var port = rest[0] === ":" ? parseInt(rest.slice(1), 10) : null;
      return [host, isNaN(port) ? null : port];
    }
    if ((h.match(/:/g) || []).length === 1) {
      var parts = h.split(":");
      var p = parseInt(parts[1], 10);
      if (!isNaN(p) && String(p) === parts[1]) return [parts[0], p];
      return [h, null];
    }
    return [h, null];
  }

  function isCloudHost(host) {
    var h = normalizeHost(host);
    return ["trycloudflare.com", "proxy.runpod.net", "ngrok"].some(
      function (t) { return h.indexOf(t) !== -1; });
  }

  function buildWorkerUrl(worker) {
    var host = normalizeHost(worker.host || "localhost");
    var port = worker.port;
    var sp = splitHostPort(host);
    if (sp[1] !== null) { host = sp[0]; port = sp[1]; }
    if (isCloudHost(host)) return "https://" + host;
    if (!host) host = "localhost";
    return "http://" + host + ":" + port;
  }

  function buildMasterUrl(master, defaultPort) {
    var host = normalizeHost((master && master.host) || "");
    var port = (master && master.port) || defaultPort || 8188;
    if (host && isCloudHost(host)) return "https://" + host;
    return "http://" + (host || "127.0.0.1") + ":" + port;
  }

  // ---- prompt-graph scanning (reference workerUtils.findNodesByClass) ----
  function findNodesByClass(prompt, className) {
    var out = [];
    Object.keys(prompt || {}).forEach(function (id) {
      var n = prompt[id];
      if (n && n.class_type === className) out.push(id);
    });
    return out;
  }

  function hasDistributedNodes(prompt) {
    return findNodesByClass(prompt, NODE_CLASSES.COLLECTOR).length > 0 ||
           findNodesByClass(prompt, NODE_CLASSES.UPSCALE).length > 0;
  }

  // ---- submission decision (reference executionUtils.js:6-151) -----------
  // probeResults: {workerId: truthy when online}; masterReachable: null =
  // not checked / not applicable, false = cloudflare tunnel dead.
  function decideSubmission(opts) {
    var prompt = opts.prompt, workers = opts.enabledWorkers || [];
    if (!hasDistributedNodes(prompt))
      return { mode: "local", activeIds: [], reason: "no distributed nodes" };
    var active = workers.filter(function (w) {
      return Boolean((opts.probeResults || {})[String(w.id)]);
    });
    if (workers.length > 0 && active.length === 0)
      return { mode: "local", activeIds: [],
               reason: "all enabled workers offline" };
    var masterHost = normalizeHost(opts.masterHost || "");
    var isCf = /\.(trycloudflare\.com|cloudflare\.dev)$/i.test(masterHost);
    if (isCf && active.length > 0 && opts.masterReachable === false)
      return { mode: "blocked", activeIds: [],
               reason: "master tunnel host " + masterHost +
                       " unreachable - workers cannot send results back" };
    return {
      mode: "distributed",
      activeIds: active.map(function (w) { return String(w.id); }),
      reason: active.length + " of " + workers.length + " workers active",
    };
  }

  // ---- DistributedValue widget model (reference distributedValue.js) -----
  function parseWorkerStore(raw) {
    try {
      var v = typeof raw === "string" ? JSON.parse(raw || "{}") : (raw || {});
      return v && typeof v === "object" && !Array.isArray(v) ? v : {};
    } catch (e) { return {}; }
  }

  function coerceValue(value, valueType) {
    if (valueType === "INT") {
      var i = parseInt(parseFloat(value), 10);
      return isNaN(i) ? value : i;
    }
    if (valueType === "FLOAT") {
      var f = parseFloat(value);
      return isNaN(f) ? value : f;
    }
    return value; // STRING / COMBO stay strings
  }

  // One row per DistributedValue node: the per-enabled-worker widget set
  // the reference renders on the node itself (1-indexed worker keys).
  function valueWidgetModel(prompt, enabledWorkers) {
    return findNodesByClass(prompt, NODE_CLASSES.VALUE).map(function (id) {
      var inputs = (prompt[id] && prompt[id].inputs) || {};
      var store = parseWorkerStore(inputs.worker_values);
      var vtype = store._type || "STRING";
      return {
        nodeId: id,
        valueType: vtype,
        defaultValue: inputs.default_value != null ?
            String(inputs.default_value) : "",
        workers: enabledWorkers.map(function (w, i) {
          var key = String(i + 1); // worker_N offset convention
          return {
            id: String(w.id),
            name: w.name || String(w.id),
            key: key,
            value: store[key] != null ? String(store[key]) : "",
          };
        }),
      };
    });
  }

  // Write edited values back into the prompt (the reference stores them in
  // the node's worker_values JSON widget the same way).
  function applyValueEdits(prompt, nodeId, defaultValue, workerEdits,
                           valueType) {
    var node = prompt[nodeId];
    if (!node || node.class_type !== NODE_CLASSES.VALUE)
      throw new Error("node " + nodeId + " is not a DistributedValue");
    var store = parseWorkerStore(node.inputs.worker_values);
    if (valueType) store._type = valueType;
    Object.keys(workerEdits || {}).forEach(function (key) {
      var v = workerEdits[key];
      if (v === "" || v == null) delete store[key];
      else store[key] = coerceValue(v, store._type || "STRING");
    });
    node.inputs.default_value = String(defaultValue);
    node.inputs.worker_values = JSON.stringify(store);
    return prompt;
  }

  // ---- batch-divider socket model (reference image_batch_divider.js) -----
  var DIVIDER_NODES = {
    ImageBatchDivider: { prefix: "batch_", type: "IMAGE" },
    AudioBatchDivider: { prefix: "audio_", type: "AUDIO" },
  };

  function dividerOutputs(classType, divideBy) {
    var cfg = DIVIDER_NODES[classType];
    if (!cfg) return null;
    var n = parseInt(divideBy, 10) || 1;
    n = Math.max(1, Math.min(10, n));
    var out = [];
    for (var i = 1; i <= n; i++)
      out.push({ name: cfg.prefix + i, type: cfg.type });
    return out;
  }

  // Dangling divider links after shrinking divide_by: the editor removes
  // the sockets; on raw prompt JSON we detect consumers of now-missing
  // outputs so the panel can warn before submission.
  function danglingDividerLinks(prompt) {
    var bad = [];
    Object.keys(prompt || {}).forEach(function (id) {
      var n = prompt[id];
      if (!n || !n.inputs) return;
      Object.keys(n.inputs).forEach(function (name) {
        var v = n.inputs[name];
        if (!Array.isArray(v) || v.length !== 2) return;
        var src = prompt[String(v[0])];
        if (!src || !DIVIDER_NODES[src.class_type]) return;
        var nOut = (dividerOutputs(src.class_type,
                                   src.inputs && src.inputs.divide_by) || [])
            .length;
        if (v[1] >= nOut)
          bad.push({ node: id, input: name, source: String(v[0]),
                     output: v[1], available: nOut });
      });
    });
    return bad;
  }

  return {
    NODE_CLASSES: NODE_CLASSES,
    normalizeHost: normalizeHost,
    splitHostPort: splitHostPort,
    isCloudHost: isCloudHost,
    buildWorkerUrl: buildWorkerUrl,
    buildMasterUrl: buildMasterUrl,
    findNodesByClass: findNodesByClass,
    hasDistributedNodes: hasDistributedNodes,
    decideSubmission: decideSubmission,
    parseWorkerStore: parseWorkerStore,
    coerceValue: coerceValue,
    valueWidgetModel: valueWidgetModel,
    applyValueEdits: applyValueEdits,
    dividerOutputs: dividerOutputs,
    danglingDividerLinks: danglingDividerLinks,
  };
});
