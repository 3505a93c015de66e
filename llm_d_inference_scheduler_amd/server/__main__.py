"""Serve CLI — boot the node's front doors (cmd/epp/main.go analog).

    python -m llm_d_inference_scheduler_amd.server [flags]

Single-rank (world=1) mode runs router + engine in this process; for
multi-GPU serving launch one process per GPU via torchrun with the same
flags (rank 0 hosts the HTTP/ext-proc front doors).

Flag surface mirrors the reference's pflag options (pkg/epp/server/
options.go): ports, pool identity via topology, the EndpointPickerConfig
YAML (--config-file/--config-text), feature toggles.
"""
import argparse
import os
import sys


def parse_args(argv=None):
    p = argparse.ArgumentParser(
        prog="llm_d_inference_scheduler_amd.server")
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=8000,
                   help="OpenAI-compatible HTTP front door")
    p.add_argument("--extproc-port", type=int, default=0,
                   help="Envoy ext-proc gRPC front door (0 = disabled; "
                        "the reference's default grpcPort is 9002)")
    p.add_argument("--model", default="llama-3-8b",
                   choices=["llama-3-8b", "qwen3-32b", "llava-1.5-7b",
                            "tiny-llama"])
    p.add_argument("--topology", default="mono",
                   help='"mono", "pd-combined", "pd:NpMd", "epd:1eNpMd"')
    p.add_argument("--config-file", default="",
                   help="EndpointPickerConfig YAML path")
    p.add_argument("--config-text", default="",
                   help="EndpointPickerConfig YAML inline")
    p.add_argument("--kv-gb", type=float, default=160.0)
    p.add_argument("--kv-dtype", default="auto",
                   choices=["auto", "bf16", "fp8"])
    p.add_argument("--flow-control", action="store_true",
                   help="enable the flowControl feature gate")
    p.add_argument("--decode-chunk-tokens", type=int, default=0)
    p.add_argument("--device", default=None)
    p.add_argument("--seed", type=int, default=0)
    return p.parse_args(argv)


def build_node(args):
    import torch

    from ..models.configs import (LLAMA_3_8B, LLAVA_1_5_7B_TEXT,
                                  QWEN3_32B, TINY_LLAMA)
    from ..node import NodeConfig, NodeRunner
    model = {"llama-3-8b": LLAMA_3_8B, "qwen3-32b": QWEN3_32B,
             "llava-1.5-7b": LLAVA_1_5_7B_TEXT,
             "tiny-llama": TINY_LLAMA}[args.model]
    use_gpu = torch.cuda.is_available() if args.device is None else \
        str(args.device).startswith("cuda")
    device = args.device or ("cuda:0" if use_gpu else "cpu")
    epp_yaml = args.config_text
    if args.config_file:
        with open(args.config_file) as f:
            epp_yaml = f.read()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        import torch.distributed as dist
        dist.init_process_group("nccl" if use_gpu else "gloo")
    cfg = NodeConfig(
        model=model, rank=rank, world_size=world, topology=args.topology,
        epp_yaml=epp_yaml, device=device,
        dtype=torch.bfloat16 if use_gpu else torch.float32,
        kv_blocks=None if use_gpu else 2048,
        kv_budget_bytes=int(args.kv_gb * (1 << 30)),
        kv_cache_dtype=args.kv_dtype,
        flow_control=args.flow_control,
        decode_chunk_tokens=args.decode_chunk_tokens or None,
        seed=args.seed)
    return NodeRunner(cfg)


def main(argv=None):
    args = parse_args(argv)
    node = build_node(args)
    if node.rank != 0:
        # worker rank: just run the lockstep loop
        try:
            while True:
                node.step()
        except KeyboardInterrupt:
            node.shutdown()
        return
    from .openai_app import build_app
    from .service import NodeService
    service = NodeService(node)
    service.start()
    extproc = None
    if args.extproc_port:
        from .extproc import ExtProcServer
        extproc = ExtProcServer(node)
        extproc.start(args.extproc_port)
    app = build_app(service)
    import uvicorn
    try:
        uvicorn.run(app, host=args.host, port=args.port, log_level="info")
    finally:
        if extproc is not None:
            extproc.stop()
        service.stop()


if __name__ == "__main__":
    sys.exit(main())
