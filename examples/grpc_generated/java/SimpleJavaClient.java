// Bare-stub Java client for the KServe-v2 gRPC service (reference:
// src/grpc_generated/java/.../SimpleJavaClient.java — reimplemented for
// this stack's vendored proto/grpc_service.proto).
//
// Generate stubs first:
//   protoc --java_out=. --plugin=protoc-gen-grpc-java \
//       --grpc-java_out=. proto/grpc_service.proto
//
// NOTE: no JVM ships in this repo's CI image; compile-checked only
// where a toolchain exists.
import java.nio.ByteBuffer;
import java.nio.ByteOrder;

import com.google.protobuf.ByteString;

import inference.GRPCInferenceServiceGrpc;
import inference.GrpcService.ModelInferRequest;
import inference.GrpcService.ModelInferResponse;
import inference.GrpcService.ServerLiveRequest;
import io.grpc.ManagedChannel;
import io.grpc.ManagedChannelBuilder;

public class SimpleJavaClient {
  public static void main(String[] args) {
    String host = args.length > 0 ? args[0] : "127.0.0.1";
    int port = args.length > 1 ? Integer.parseInt(args[1]) : 8001;
    ManagedChannel channel = ManagedChannelBuilder
        .forAddress(host, port).usePlaintext().build();
    GRPCInferenceServiceGrpc.GRPCInferenceServiceBlockingStub stub =
        GRPCInferenceServiceGrpc.newBlockingStub(channel);

    if (!stub.serverLive(ServerLiveRequest.newBuilder().build()).getLive()) {
      throw new RuntimeException("server not live");
    }

    ByteBuffer in0 = ByteBuffer.allocate(64).order(ByteOrder.LITTLE_ENDIAN);
    ByteBuffer in1 = ByteBuffer.allocate(64).order(ByteOrder.LITTLE_ENDIAN);
    for (int i = 0; i < 16; i++) { in0.putInt(i); in1.putInt(1); }

    ModelInferRequest request = ModelInferRequest.newBuilder()
        .setModelName("simple")
        .addInputs(ModelInferRequest.InferInputTensor.newBuilder()
            .setName("INPUT0").setDatatype("INT32")
            .addShape(1).addShape(16))
        .addInputs(ModelInferRequest.InferInputTensor.newBuilder()
            .setName("INPUT1").setDatatype("INT32")
            .addShape(1).addShape(16))
        .addRawInputContents(ByteString.copyFrom(in0.array()))
        .addRawInputContents(ByteString.copyFrom(in1.array()))
        .build();

    ModelInferResponse response = stub.modelInfer(request);
    ByteBuffer out0 = response.getRawOutputContents(0).asReadOnlyByteBuffer()
        .order(ByteOrder.LITTLE_ENDIAN);
    for (int i = 0; i < 16; i++) {
      if (out0.getInt(i * 4) != i + 1) {
        throw new RuntimeException("mismatch at " + i);
      }
    }
    System.out.println("PASS: java bare-stub client");
    channel.shutdown();
  }
}
