// Bare-stub Go client for the KServe-v2 gRPC service (reference:
// src/grpc_generated/go/grpc_simple_client.go — reimplemented for this
// stack's vendored proto/grpc_service.proto).
//
// Generate stubs first (writes the inference package):
//   protoc --go_out=. --go-grpc_out=. ../../proto/grpc_service.proto
// Then:
//   go run grpc_simple_client.go -u 127.0.0.1:8001
//
// NOTE: no Go toolchain ships in this repo's CI image; this sample is
// compile-checked only where a toolchain exists.
package main

import (
	"context"
	"encoding/binary"
	"flag"
	"fmt"
	"log"
	"time"

	"google.golang.org/grpc"
	"google.golang.org/grpc/credentials/insecure"

	pb "client_amd_example/inference" // from protoc --go_out
)

func int32Bytes(vals []int32) []byte {
	buf := make([]byte, 4*len(vals))
	for i, v := range vals {
		binary.LittleEndian.PutUint32(buf[i*4:], uint32(v))
	}
	return buf
}

func main() {
	url := flag.String("u", "127.0.0.1:8001", "server URL")
	flag.Parse()

	conn, err := grpc.NewClient(
		*url, grpc.WithTransportCredentials(insecure.NewCredentials()))
	if err != nil {
		log.Fatalf("connect: %v", err)
	}
	defer conn.Close()
	client := pb.NewGRPCInferenceServiceClient(conn)
	ctx, cancel := context.WithTimeout(context.Background(), 10*time.Second)
	defer cancel()

	live, err := client.ServerLive(ctx, &pb.ServerLiveRequest{})
	if err != nil || !live.Live {
		log.Fatalf("server not live: %v", err)
	}

	in0 := make([]int32, 16)
	in1 := make([]int32, 16)
	for i := range in0 {
		in0[i] = int32(i)
		in1[i] = 1
	}
	req := &pb.ModelInferRequest{
		ModelName: "simple",
		Inputs: []*pb.ModelInferRequest_InferInputTensor{
			{Name: "INPUT0", Datatype: "INT32", Shape: []int64{1, 16}},
			{Name: "INPUT1", Datatype: "INT32", Shape: []int64{1, 16}},
		},
		RawInputContents: [][]byte{int32Bytes(in0), int32Bytes(in1)},
	}
	resp, err := client.ModelInfer(ctx, req)
	if err != nil {
		log.Fatalf("infer: %v", err)
	}
	out0 := resp.RawOutputContents[0]
	for i := 0; i < 16; i++ {
		sum := int32(binary.LittleEndian.Uint32(out0[i*4:]))
		if sum != in0[i]+in1[i] {
			log.Fatalf("mismatch at %d: %d", i, sum)
		}
	}
	fmt.Println("PASS: go bare-stub client")
}
