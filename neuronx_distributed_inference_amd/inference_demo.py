"""`inference_demo` CLI.

Analog of the reference's `inference_demo.py` (:73-803): argparse front-end
over the compile -> load -> accuracy-check -> generate -> benchmark
pipeline, with the same flag names where they exist on MI355X. Multi-rank
runs launch this same entry under torchrun (one process per GPU over RCCL).

    inference_demo --model-type llama --task-type causal-lm run \
        --model-path /ckpt --compiled-model-path /compiled \
        --torch-dtype bfloat16 --tp-degree 8 --batch-size 2 \
        --max-context-length 128 --seq-len 256 \
        --on-device-sampling --top-k 1 \
        --prompt "I believe the meaning of life is" \
        --check-accuracy-mode token-matching --benchmark
"""

from __future__ import annotations

import argparse
import json
import logging
import os
import sys
from typing import List, Optional

import torch

from neuronx_distributed_inference_amd.config import (
    EngineConfig,
    MoEEngineConfig,
    OnDeviceSamplingConfig,
)
from neuronx_distributed_inference_amd.registry import available_model_types, get_model_classes

logger = logging.getLogger("nxdi_amd")

MODEL_TYPES = available_model_types()
CHECK_ACCURACY_MODES = ["skip-accuracy-check", "token-matching", "logit-matching"]


def parse_args(argv=None):
    p = argparse.ArgumentParser(prog="inference_demo")
    p.add_argument("--model-type", required=True, choices=MODEL_TYPES)
    p.add_argument("--task-type", default="causal-lm", choices=["causal-lm"])
    sub = p.add_subparsers(dest="command", required=True)
    run = sub.add_parser("run")
    setup_run_parser(run)
    return p.parse_args(argv)


def setup_run_parser(run: argparse.ArgumentParser):
    run.add_argument("--model-path", required=True)
    run.add_argument("--compiled-model-path", required=True)
    # generation
    run.add_argument("--prompt", action="append", dest="prompts", default=None)
    run.add_argument("--prompt-ids", default=None,
                     help="JSON list-of-lists of token ids (tokenizer-free runs)")
    run.add_argument("--max-new-tokens", type=int, default=32)
    run.add_argument("--top-k", type=int, default=1)
    run.add_argument("--top-p", type=float, default=1.0)
    run.add_argument("--temperature", type=float, default=1.0)
    run.add_argument("--do-sample", action="store_true")
    run.add_argument("--global-topk", type=int, default=256)
    run.add_argument("--on-device-sampling", action="store_true")
    run.add_argument("--pad-token-id", type=int, default=0)
    # engine geometry
    run.add_argument("--torch-dtype", default="bfloat16")
    run.add_argument("--batch-size", type=int, default=1)
    run.add_argument("--max-batch-size", type=int, default=None)
    run.add_argument("--ctx-batch-size", type=int, default=None)
    run.add_argument("--tkg-batch-size", type=int, default=None)
    run.add_argument("--max-context-length", type=int, default=128)
    run.add_argument("--seq-len", type=int, default=256)
    run.add_argument("--n-active-tokens", type=int, default=1)
    run.add_argument("--enable-bucketing", action="store_true", default=True)
    run.add_argument("--no-bucketing", dest="enable_bucketing", action="store_false")
    run.add_argument("--context-encoding-buckets", nargs="+", type=int, default=None)
    run.add_argument("--token-generation-buckets", nargs="+", type=int, default=None)
    run.add_argument("--continuous-batching", action="store_true")
    # parallelism
    run.add_argument("--tp-degree", type=int, default=1)
    run.add_argument("--pp-degree", type=int, default=1)
    run.add_argument("--ep-degree", type=int, default=1)
    run.add_argument("--cp-degree", type=int, default=1)
    run.add_argument("--attention-dp-degree", type=int, default=1)
    run.add_argument("--start-rank-id", type=int, default=0)
    run.add_argument("--local-ranks-size", type=int, default=None)
    # speculation
    run.add_argument("--draft-model-path", default=None)
    run.add_argument("--draft-model-type", default=None,
                     help="model family of the draft (defaults to the target's)")
    run.add_argument("--draft-tp-degree", type=int, default=None,
                     help="run the draft at a smaller TP (sub-communicator; "
                          "must divide --tp-degree)")
    run.add_argument("--speculation-length", type=int, default=0)
    run.add_argument("--enable-fused-speculation", action="store_true")
    run.add_argument("--spec-rejection-sampling", action="store_true",
                     help="true rejection sampling in the fused round "
                          "(spec decode under top-k/top-p sampling)")
    run.add_argument("--medusa-speculation-length", type=int, default=0)
    # engine knobs
    run.add_argument("--no-hip-graphs", dest="enable_hip_graphs", action="store_false",
                     default=True)
    run.add_argument("--kv-cache-quant", action="store_true",
                     help="store the KV cache in fp8 (OCP e4m3)")
    run.add_argument("--quantized", action="store_true",
                     help="fp8 weight quantization (per-row scales)")
    run.add_argument("--moe-mxfp4", action="store_true",
                     help="keep GPT-OSS MoE weights in MXFP4 packing "
                          "(in-kernel dequant)")
    run.add_argument("--fused-qkv", action="store_true")
    run.add_argument("--fused-gate-up", action="store_true")
    run.add_argument("--sequence-parallel-enabled", action="store_true")
    run.add_argument("--windowed-context-encoding-size", type=int, default=0)
    run.add_argument("--is-block-kv-layout", action="store_true")
    run.add_argument("--pa-num-blocks", type=int, default=None)
    run.add_argument("--pa-block-size", type=int, default=128)
    run.add_argument("--skip-warmup", action="store_true")
    run.add_argument("--on-cpu", action="store_true")
    # extended parity flags (ref inference_demo.py:73-435 surface)
    run.add_argument("--async-mode", action="store_true",
                     help="async double-buffered serving decode")
    run.add_argument("--output-logits", action="store_true")
    run.add_argument("--flash-decoding", dest="flash_decoding_enabled",
                     action="store_true")
    run.add_argument("--num-cores-per-group", type=int, default=None)
    run.add_argument("--mlp-cp-degree", type=int, default=1)
    run.add_argument("--sampling-dp-degree", type=int, default=1)
    run.add_argument("--moe-tp-degree", type=int, default=None)
    run.add_argument("--moe-ep-degree", type=int, default=None)
    run.add_argument("--vocab-parallel", action="store_true")
    run.add_argument("--compact-swa-kv", action="store_true",
                     help="ring (compact) KV cache for sliding-window layers")
    run.add_argument("--is-prefix-caching", action="store_true")
    run.add_argument("--is-chunked-prefill", action="store_true")
    run.add_argument("--quantize-moe", choices=["fp8", "mx4"], default=None,
                     help="runtime-quantize MoE expert stacks (fp8-e4m3 "
                          "per-channel, or OCP MXFP4 at 4.25 bits)")
    run.add_argument("--enable-eagle-speculation", action="store_true")
    run.add_argument("--enable-token-tree", action="store_true",
                     help="EAGLE TREE speculation (needs --enable-eagle-speculation)")
    run.add_argument("--token-tree-branching", default="2,2",
                     help="per-level branching factors of the static draft tree")
    run.add_argument("--enable-eagle3", action="store_true",
                     help="EAGLE-3: draft conditioned on 3 target layers' hiddens")
    run.add_argument("--eagle3-capture-layers", default=None,
                     help="csv of target layer indices (default low/mid/high)")
    run.add_argument("--num-medusa-heads", type=int, default=0)
    run.add_argument("--quantization-type", default=None,
                     choices=[None, "per_tensor_symmetric", "per_channel_symmetric"])
    run.add_argument("--quantization-dtype", default=None,
                     choices=[None, "f8e4m3", "int8"])
    run.add_argument("--rpl-reduce-dtype", default=None)
    run.add_argument("--attention-dtype", default=None)
    run.add_argument("--cast-type", default="config", choices=["config", "as-declared"])
    run.add_argument("--no-batch-bucketing", dest="enable_batch_bucketing",
                     action="store_false", default=True)
    run.add_argument("--token-generation-batch-buckets", nargs="+", type=int,
                     default=None)
    run.add_argument("--padding-side", default="right", choices=["right", "left"])
    run.add_argument("--lora-paths", nargs="+", default=None,
                     help="name=path LoRA adapter dirs to preload")
    run.add_argument("--max-loras", type=int, default=4)
    run.add_argument("--max-lora-rank", type=int, default=16)
    # pipeline control
    run.add_argument("--check-accuracy-mode", default="skip-accuracy-check",
                     choices=CHECK_ACCURACY_MODES)
    run.add_argument("--divergence-difference-tol", type=float, default=0.001)
    run.add_argument("--capture-inputs-on-divergence", action="store_true",
                     help="on logit divergence, re-run with tensor capture "
                          "on every decoder layer and dump the tensors at "
                          "the divergence step (ref inference_demo.py:616-649)")
    run.add_argument("--capture-modules", nargs="+", default=None,
                     help="module names to capture (default: every layer)")
    run.add_argument("--benchmark", action="store_true")
    run.add_argument("--num-benchmark-runs", type=int, default=5)
    run.add_argument("--skip-compile", action="store_true")
    run.add_argument("--compile-only", action="store_true")
    run.add_argument("--seed", type=int, default=0)


def create_engine_config(args, model_type: str) -> EngineConfig:
    """args -> EngineConfig (ref create_neuron_config, inference_demo.py:436)."""
    odsc = None
    if args.on_device_sampling:
        odsc = OnDeviceSamplingConfig(
            top_k=args.top_k, top_p=args.top_p, temperature=args.temperature,
            do_sample=args.do_sample, global_topk=args.global_topk,
        )
    elif getattr(args, "enable_eagle_speculation", False):
        # EAGLE verification consumes on-device greedy tokens (the ref
        # likewise forces on-device sampling under speculation).
        odsc = OnDeviceSamplingConfig(top_k=1)
    cls = MoEEngineConfig if model_type in ("mixtral", "dbrx", "qwen3_moe") else EngineConfig
    return cls(
        batch_size=args.batch_size,
        max_batch_size=args.max_batch_size or args.batch_size,
        ctx_batch_size=args.ctx_batch_size,
        tkg_batch_size=args.tkg_batch_size,
        max_context_length=args.max_context_length,
        seq_len=args.seq_len,
        max_length=args.seq_len,
        n_active_tokens=args.n_active_tokens,
        torch_dtype=args.torch_dtype if not args.on_cpu else "float32",
        enable_bucketing=args.enable_bucketing,
        context_encoding_buckets=args.context_encoding_buckets,
        token_generation_buckets=args.token_generation_buckets,
        is_continuous_batching=args.continuous_batching,
        tp_degree=args.tp_degree,
        pp_degree=args.pp_degree,
        ep_degree=args.ep_degree,
        cp_degree=args.cp_degree,
        attention_dp_degree=args.attention_dp_degree,
        start_rank_id=args.start_rank_id,
        local_ranks_size=args.local_ranks_size,
        on_device_sampling_config=odsc,
        speculation_length=args.speculation_length,
        enable_fused_speculation=args.enable_fused_speculation,
        enable_hip_graphs=args.enable_hip_graphs,
        skip_warmup=args.skip_warmup,
        on_cpu=args.on_cpu,
        kv_cache_quant=args.kv_cache_quant,
        moe_mxfp4=getattr(args, "moe_mxfp4", False),
        quantized=args.quantized,
        fused_qkv=args.fused_qkv,
        fused_gate_up=args.fused_gate_up,
        sequence_parallel_enabled=args.sequence_parallel_enabled,
        windowed_context_encoding_size=args.windowed_context_encoding_size,
        is_block_kv_layout=args.is_block_kv_layout,
        pa_num_blocks=args.pa_num_blocks,
        pa_block_size=args.pa_block_size,
        async_mode=getattr(args, "async_mode", False),
        output_logits=getattr(args, "output_logits", False),
        flash_decoding_enabled=getattr(args, "flash_decoding_enabled", False),
        num_cores_per_group=getattr(args, "num_cores_per_group", None),
        mlp_cp_degree=getattr(args, "mlp_cp_degree", 1),
        sampling_dp_degree=getattr(args, "sampling_dp_degree", 1),
        moe_tp_degree=getattr(args, "moe_tp_degree", None),
        moe_ep_degree=getattr(args, "moe_ep_degree", None),
        vocab_parallel=getattr(args, "vocab_parallel", False),
        compact_swa_kv=getattr(args, "compact_swa_kv", False),
        is_prefix_caching=getattr(args, "is_prefix_caching", False),
        is_chunked_prefill=getattr(args, "is_chunked_prefill", False),
        enable_eagle_speculation=getattr(args, "enable_eagle_speculation", False),
        num_medusa_heads=getattr(args, "num_medusa_heads", 0),
        is_medusa=getattr(args, "num_medusa_heads", 0) > 0,
        medusa_speculation_length=getattr(args, "medusa_speculation_length", 0),
        quantization_type=getattr(args, "quantization_type", None),
        quantization_dtype=getattr(args, "quantization_dtype", None),
        cast_type=getattr(args, "cast_type", "config"),
        enable_batch_bucketing=getattr(args, "enable_batch_bucketing", True),
        token_generation_batch_buckets=getattr(
            args, "token_generation_batch_buckets", None),
        padding_side=getattr(args, "padding_side", "right"),
        **({"rpl_reduce_dtype": args.rpl_reduce_dtype}
           if getattr(args, "rpl_reduce_dtype", None) else {}),
        **({"attention_dtype": args.attention_dtype}
           if getattr(args, "attention_dtype", None) else {}),
    )


def load_tokenizer(path: str):
    try:
        from transformers import AutoTokenizer

        return AutoTokenizer.from_pretrained(path)
    except Exception as e:  # noqa: BLE001
        logger.warning("no tokenizer loaded from %s (%s)", path, e)
        return None


def get_input_ids(args, tokenizer, device):
    if args.prompt_ids:
        ids = json.loads(args.prompt_ids)
        maxlen = max(len(r) for r in ids)
        input_ids = torch.full((len(ids), maxlen), args.pad_token_id, dtype=torch.long)
        mask = torch.zeros_like(input_ids)
        for i, r in enumerate(ids):
            input_ids[i, : len(r)] = torch.tensor(r)
            mask[i, : len(r)] = 1
        return input_ids.to(device), mask.to(device)
    assert tokenizer is not None, "--prompt requires tokenizer files beside the model"
    prompts = args.prompts or ["I believe the meaning of life is"]
    enc = tokenizer(prompts, return_tensors="pt", padding=True)
    return enc["input_ids"].to(device), enc["attention_mask"].to(device)


def run_inference(args, model_type: str) -> int:
    from neuronx_distributed_inference_amd.models.application_base import (
        InferenceApplication,
    )
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    torch.manual_seed(args.seed)
    config_cls, model_cls = get_model_classes(model_type)
    if not hasattr(model_cls, "_model_cls"):
        print(f"model type {model_type!r} is a conditional-generation family "
              "(speech / vision); drive it programmatically via "
              f"{model_cls.__module__} — the CLI run path covers causal LMs.")
        return 2
    local_rank = ps.init_distributed_from_env()
    ps.initialize_model_parallel(
        tensor_model_parallel_size=args.tp_degree,
        pipeline_model_parallel_size=args.pp_degree,
        expert_model_parallel_size=args.ep_degree,
        context_parallel_size=args.cp_degree,
        attention_dp_size=args.attention_dp_degree,
        draft_tp_size=getattr(args, "draft_tp_degree", None),
    )
    device = torch.device(f"cuda:{local_rank}") if (
        torch.cuda.is_available() and not args.on_cpu
    ) else torch.device("cpu")

    engine_config = create_engine_config(args, model_type)
    config = config_cls.from_hf_config(args.model_path, neuron_config=engine_config)
    if getattr(args, "enable_eagle3", False):
        # resolve capture layers BEFORE load() so graph capture records them
        if args.eagle3_capture_layers:
            caps = [int(x) for x in args.eagle3_capture_layers.split(",")]
        else:
            n = config.num_hidden_layers
            caps = [1, n // 2, n - 2] if n >= 4 else [0, 0, 0]
        engine_config.eagle3_capture_layers = caps
    app = InferenceApplication(model_cls, config, model_path=args.model_path)

    rank0 = ps.get_tensor_model_parallel_rank() == 0
    if not args.skip_compile:
        if rank0:
            app.compile(args.compiled_model_path)
        if torch.distributed.is_initialized():
            torch.distributed.barrier()
    if args.compile_only:
        return 0

    model = app.load(args.compiled_model_path, device=device)
    if getattr(args, "quantize_moe", None):
        from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

        nq = 0
        for mod in model.model.modules():
            if isinstance(mod, ExpertMLPs):
                ok = (mod.quantize_fp8() if args.quantize_moe == "fp8"
                      else mod.quantize_mx4())
                nq += int(ok)
        if rank0:
            logger.info("quantized %d expert stacks to %s", nq,
                        args.quantize_moe)
    tokenizer = load_tokenizer(args.compiled_model_path) or load_tokenizer(args.model_path)
    input_ids, attention_mask = get_input_ids(args, tokenizer, device)

    gc = GenerationConfig(
        max_new_tokens=args.max_new_tokens, do_sample=args.do_sample,
        top_k=args.top_k, top_p=args.top_p, temperature=args.temperature,
        pad_token_id=args.pad_token_id,
        eos_token_id=getattr(tokenizer, "eos_token_id", None) if tokenizer else None,
    )

    if args.check_accuracy_mode != "skip-accuracy-check" and rank0:
        run_accuracy_check(args, model, config, input_ids, attention_mask, gc)

    if args.draft_model_path and args.speculation_length > 0:
        # two-model speculation (ref speculation_model submodel /
        # NeuronFusedSpecModel; draft class resolved like the reference's
        # dotted-path reconstruction, config.py:956-1038)
        from neuronx_distributed_inference_amd.modules.speculation import (
            FusedSpeculativeAdapter,
            SpeculativeGenerationAdapter,
        )

        d_cfg_cls, d_model_cls = get_model_classes(
            args.draft_model_type or model_type)
        draft_ec = create_engine_config(args, args.draft_model_type or model_type)
        draft_ec.speculation_length = 0
        draft_tp = getattr(args, "draft_tp_degree", None)
        if draft_tp and draft_tp < args.tp_degree:
            draft_ec.tp_degree = draft_tp
        draft_cfg = d_cfg_cls.from_hf_config(
            args.draft_model_path, neuron_config=draft_ec)
        draft_app = InferenceApplication(
            d_model_cls, draft_cfg, model_path=args.draft_model_path)
        draft_dir = os.path.join(args.compiled_model_path, "draft")
        if not args.skip_compile and rank0:
            draft_app.compile(draft_dir)
        if torch.distributed.is_initialized():
            torch.distributed.barrier()
        if draft_tp and draft_tp < args.tp_degree:
            # build/shard/run the draft over the reduced-TP subgroup
            # (sharding math is config-driven; only the collectives need
            # the group override)
            from neuronx_distributed_inference_amd.modules.speculation import (
                DraftTPWrapper,
            )

            with ps.draft_tp_context():
                draft = DraftTPWrapper(draft_app.load(draft_dir, device=device))
        else:
            draft = draft_app.load(draft_dir, device=device)
        if args.enable_eagle_speculation:
            # EAGLE drafts wrap the draft DecoderModel with the fc combiner;
            # fc weights come from the draft checkpoint when present
            # (ref eagle draft fc, modeling_llama.py:1161-1167), else
            # random-init (synthetic demos).
            from neuronx_distributed_inference_amd.modules.checkpoint import (
                load_state_dict,
            )
            from neuronx_distributed_inference_amd.modules.eagle import (
                Eagle3DraftModel,
                Eagle3GenerationAdapter,
                EagleDraftModel,
                EagleGenerationAdapter,
                EagleTreeAdapter,
            )

            hid = config.hidden_size
            dt = draft_ec.torch_dtype
            if isinstance(dt, str):
                dt = getattr(torch, dt)
            if args.enable_eagle3:
                edraft = Eagle3DraftModel(
                    draft.model, hid, dt, engine_config.eagle3_capture_layers)
            else:
                edraft = EagleDraftModel(draft.model, hid, dt)
            try:
                sd = load_state_dict(args.draft_model_path)
                fc_w = sd.get("fc.weight", sd.get("model.fc.weight"))
                if fc_w is not None and fc_w.shape == edraft.fc.weight.shape:
                    with torch.no_grad():
                        edraft.fc.weight.copy_(fc_w)
            except (OSError, ValueError):
                pass
            edraft = edraft.to(device)
            if args.enable_eagle3:
                adapter = Eagle3GenerationAdapter(
                    model, edraft, args.speculation_length)
            elif args.enable_token_tree:
                from neuronx_distributed_inference_amd.modules.token_tree import (
                    TokenTree,
                )

                branching = [int(x) for x in args.token_tree_branching.split(",")]
                adapter = EagleTreeAdapter(model, edraft, TokenTree(branching))
            else:
                adapter = EagleGenerationAdapter(
                    model, edraft, args.speculation_length)
        elif args.enable_fused_speculation:
            rej = None
            if args.spec_rejection_sampling:
                rej = engine_config.on_device_sampling_config or \
                    OnDeviceSamplingConfig(
                        top_k=args.top_k, top_p=args.top_p,
                        temperature=args.temperature,
                        do_sample=args.do_sample,
                        global_topk=args.global_topk)
            adapter = FusedSpeculativeAdapter(
                model, draft, args.speculation_length, rejection_config=rej)
        else:
            adapter = SpeculativeGenerationAdapter(
                model, draft, args.speculation_length)
        out = adapter.generate(
            input_ids, attention_mask=attention_mask, generation_config=gc)
    else:
        out = HuggingFaceGenerationAdapter(model).generate(
            input_ids, attention_mask=attention_mask, generation_config=gc
        )
    if rank0:
        if tokenizer is not None:
            for i, row in enumerate(out):
                print(f"--- generation {i} ---")
                print(tokenizer.decode(row, skip_special_tokens=True))
        else:
            print(out.tolist())

    if args.benchmark and rank0:
        from neuronx_distributed_inference_amd.utils.benchmark import benchmark_sampling

        report = benchmark_sampling(
            model, input_ids, attention_mask=attention_mask,
            n_runs=args.num_benchmark_runs, generation_config=gc,
            report_path=os.path.join(args.compiled_model_path, "benchmark_report.json"),
        )
        print(json.dumps(report, indent=2))
    return 0


def run_accuracy_check(args, model, config, input_ids, attention_mask, gc) -> None:
    """Token/logit matching vs the HF CPU implementation of the same
    checkpoint, with optional tensor capture at the divergence step
    (ref run_accuracy_check :710 + capture-on-divergence :616-649)."""
    from transformers import AutoModelForCausalLM

    from neuronx_distributed_inference_amd.utils.accuracy import (
        LogitMatchingValidationError,
        check_accuracy,
        check_accuracy_logits,
    )

    hf = AutoModelForCausalLM.from_pretrained(args.model_path, torch_dtype=torch.float32)
    hf.eval()
    if args.check_accuracy_mode == "token-matching":
        check_accuracy(
            model, input_ids, attention_mask=attention_mask, hf_model=hf,
            generation_config=gc,
        )
        print("accuracy check passed (token matching)")
        return
    # logit matching: HF greedy loop produces per-position golden logits
    golden = compute_hf_golden_logits(hf, input_ids, attention_mask,
                                      gc.max_new_tokens)
    tol = args.divergence_difference_tol
    try:
        check_accuracy_logits(model, input_ids, golden,
                              attention_mask=attention_mask,
                              atol=tol, rtol=tol)
        print("accuracy check passed (logit matching)")
    except LogitMatchingValidationError as e:
        if args.capture_inputs_on_divergence:
            from neuronx_distributed_inference_amd.utils.tensor_capture import (
                TensorCapture,
            )

            names = args.capture_modules or [
                n for n, _ in model.model.named_modules()
                if n.startswith("layers.") and n.count(".") == 1
            ]
            model.reset()
            with TensorCapture(model.model, names,
                               max_steps=e.divergence_index + 1) as cap:
                try:
                    check_accuracy_logits(model, input_ids, golden,
                                          attention_mask=attention_mask,
                                          atol=tol, rtol=tol)
                except LogitMatchingValidationError:
                    pass
            dump = os.path.join(args.compiled_model_path,
                                f"divergence_capture_idx{e.divergence_index}.pt")
            cap.save(dump)
            logger.error("logit divergence at %d; captured %d modules -> %s",
                         e.divergence_index, len(names), dump)
        raise


@torch.inference_mode()
def compute_hf_golden_logits(hf, input_ids, attention_mask, n_positions: int):
    """HF greedy loop collecting the last-token logits at every position
    (the golden tensor for logit matching)."""
    out = []
    ids = input_ids.clone()
    mask = (attention_mask.clone() if attention_mask is not None
            else torch.ones_like(ids))
    for _ in range(n_positions):
        logits = hf(input_ids=ids, attention_mask=mask).logits[:, -1, :].float()
        out.append(logits)
        tok = logits.argmax(-1, keepdim=True)
        ids = torch.cat([ids, tok], dim=-1)
        mask = torch.cat([mask, torch.ones_like(tok)], dim=-1)
    return torch.stack(out)


def main(argv=None) -> int:
    logging.basicConfig(level=os.environ.get("NXDI_LOG_LEVEL", "INFO"))
    args = parse_args(argv)
    return run_inference(args, args.model_type)


if __name__ == "__main__":
    sys.exit(main())
