// Bare-stub Node.js client for the KServe-v2 gRPC service (reference:
// src/grpc_generated/javascript/client.js — reimplemented against this
// stack's vendored proto). Uses dynamic proto loading, so no codegen
// step is needed:
//   npm install @grpc/grpc-js @grpc/proto-loader
//   node client.js 127.0.0.1:8001
//
// NOTE: no Node toolchain ships in this repo's CI image; this sample is
// run-checked only where one exists.
const grpc = require("@grpc/grpc-js");
const protoLoader = require("@grpc/proto-loader");
const path = require("path");

const url = process.argv[2] || "127.0.0.1:8001";
const protoPath = path.join(__dirname, "..", "..", "..", "proto",
                            "grpc_service.proto");

const def = protoLoader.loadSync(protoPath, {
  keepCase: true, longs: Number, enums: String, defaults: true,
});
const inference = grpc.loadPackageDefinition(def).inference;
const client = new inference.GRPCInferenceService(
  url, grpc.credentials.createInsecure());

function int32Bytes(vals) {
  const buf = Buffer.alloc(4 * vals.length);
  vals.forEach((v, i) => buf.writeInt32LE(v, i * 4));
  return buf;
}

client.ServerLive({}, (err, res) => {
  if (err || !res.live) throw new Error(`server not live: ${err}`);
  const in0 = Array.from({length: 16}, (_, i) => i);
  const in1 = Array.from({length: 16}, () => 1);
  const request = {
    model_name: "simple",
    inputs: [
      {name: "INPUT0", datatype: "INT32", shape: [1, 16]},
      {name: "INPUT1", datatype: "INT32", shape: [1, 16]},
    ],
    raw_input_contents: [int32Bytes(in0), int32Bytes(in1)],
  };
  client.ModelInfer(request, (err2, response) => {
    if (err2) throw err2;
    const out0 = response.raw_output_contents[0];
    for (let i = 0; i < 16; i++) {
      if (out0.readInt32LE(i * 4) !== in0[i] + in1[i]) {
        throw new Error(`mismatch at ${i}`);
      }
    }
    console.log("PASS: javascript bare-stub client");
  });
});
