"""Aux subsystem tests: fp8 quant, proofs, watchdog, state keeper, config,
CLI planner, formatter."""

import json
import os

import torch

from tensorlink_amd.engine.formatter import (ResponseFormatter,
                                             extract_reasoning_and_answer,
                                             format_chat_prompt,
                                             normalize_generate_args)


def test_normalize_generate_args_clamps():
    out = normalize_generate_args({"max_new_tokens": 99999,
                                   "temperature": 9.0, "top_p": 2.0})
    assert out["max_new_tokens"] == 2048
    assert out["temperature"] == 2.0
    assert out["top_p"] == 1.0
    out = normalize_generate_args({"temperature": -5, "max_new_tokens": "x"})
    assert out["temperature"] == 0.01
    assert out["max_new_tokens"] == 256
    # beam + sample conflict resolves to sampling
    out = normalize_generate_args({"num_beams": 4, "do_sample": True})
    assert out["num_beams"] == 1
    out = normalize_generate_args({"do_sample": False})
    assert out["temperature"] == 0.0


def test_chat_prompt_templates():
    msgs = [{"role": "system", "content": "s"},
            {"role": "user", "content": "hi"}]
    qwen = format_chat_prompt(msgs, None, "Qwen/Qwen2.5-7B")
    assert "<|im_start|>system" in qwen and qwen.endswith(
        "<|im_start|>assistant\n")
    llama = format_chat_prompt(msgs, None, "meta-llama/Llama-3-8B")
    assert "<|start_header_id|>user<|end_header_id|>" in llama


def test_reasoning_extraction():
    think, ans = extract_reasoning_and_answer(
        "<think>step by step</think>The answer is 4.")
    assert think == "step by step"
    assert ans == "The answer is 4."
    think, ans = extract_reasoning_and_answer("plain")
    assert think == "" and ans == "plain"


def test_sse_chunk_shapes():
    f = ResponseFormatter("m", "openai")
    chunk = f.format_stream_chunk("hi", first=True)
    assert chunk.startswith("data: ")
    payload = json.loads(chunk[6:])
    assert payload["choices"][0]["delta"]["role"] == "assistant"
    final = f.format_final_chunk()
    assert final.endswith("data: [DONE]\n\n")


def test_fp8_quant_roundtrip():
    from tensorlink_amd.models.quant import (Fp8Linear,
                                             quantize_fp8_per_channel)
    torch.manual_seed(0)
    w = torch.randn(64, 32) * 3
    w8, scale = quantize_fp8_per_channel(w)
    assert w8.dtype == torch.float8_e4m3fn
    deq = w8.float() * scale[:, None]
    rel = (deq - w).abs().max() / w.abs().max()
    assert rel < 0.05
    lin = torch.nn.Linear(32, 64)
    f8 = Fp8Linear.from_linear(lin)
    x = torch.randn(4, 32)
    torch.testing.assert_close(f8(x), lin(x), atol=0.1, rtol=0.1)


def test_fp8_expert_conversion_and_forward():
    from tensorlink_amd.models import build_full_model, get_config
    from tensorlink_amd.models.loader import init_random_stage
    from tensorlink_amd.models.quant import Fp8Linear, quantize_experts_fp8
    cfg = get_config("tiny-moe")
    m = build_full_model(cfg)
    init_random_stage(m, dtype=torch.float32)
    n = quantize_experts_fp8(m)
    assert n == cfg.num_local_experts * 2 * cfg.num_hidden_layers
    assert isinstance(m.layers[0].mlp.experts[0].gate_up_proj, Fp8Linear)
    ids = torch.randint(0, cfg.vocab_size, (2, 8))
    pos = torch.arange(8).unsqueeze(0).expand(2, -1).contiguous()
    out = m(ids, pos)
    assert torch.isfinite(out.float()).all()


def test_gradient_hash_and_proofs():
    from tensorlink_amd.utils.proofs import (gradient_hash,
                                             verify_loss_trajectory)
    g1 = [torch.ones(4), torch.zeros(3)]
    g2 = [torch.ones(4), torch.zeros(3)]
    g3 = [torch.ones(4) * 2, torch.zeros(3)]
    assert gradient_hash(g1) == gradient_hash(g2)
    assert gradient_hash(g1) != gradient_hash(g3)
    assert verify_loss_trajectory(list(range(30, 0, -1)))
    assert not verify_loss_trajectory([1.0] * 10 + [5.0] * 20)


def test_watchdog_restarts_failed_job():
    from tensorlink_amd.engine.engine import InferenceEngine, ModelJob
    from tensorlink_amd.utils.watchdog import Watchdog
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.load_model("tiny")
    # simulate a failure
    eng.jobs["tiny"] = ModelJob(name="tiny", runner=None, tokenizer=None,
                                state="failed")
    wd = Watchdog(eng, interval_s=999)
    status = wd.check_once()
    assert wd.restarts == 1
    assert eng.jobs["tiny"].state == "ready"
    assert status["status"] in ("ok", "degraded")


def test_state_keeper_roundtrip(tmp_path):
    from tensorlink_amd.engine.engine import InferenceEngine
    from tensorlink_amd.utils.state import StateKeeper
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.demand["tiny"] = 5
    path = str(tmp_path / "state.json")
    keeper = StateKeeper(eng, path=path)
    keeper.write_state()
    eng2 = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    keeper2 = StateKeeper(eng2, path=path)
    snap = keeper2.load_previous_state()
    assert snap is not None
    assert eng2.demand["tiny"] == 5


def test_engine_config_load(tmp_path):
    from tensorlink_amd.config import EngineConfig
    p = tmp_path / "config.json"
    p.write_text(json.dumps({"node": {"endpoint_port": 9123},
                             "ml": {"default_models": ["tiny"]}}))
    cfg = EngineConfig.load(str(p))
    assert cfg.node.endpoint_port == 9123
    assert cfg.ml.default_models == ["tiny"]


def test_cli_plan(capsys):
    import tlctl
    import sys
    argv = sys.argv
    sys.argv = ["tlctl", "plan", "--model", "Qwen/Qwen3-8B", "--gpus", "4",
                "--pp", "--training"]
    try:
        tlctl.main()
    finally:
        sys.argv = argv
    out = capsys.readouterr().out
    assert "PP=4" in out and "training=True" in out


def test_tp_shard_state_partitions_exactly():
    """TP sharding: the two ranks' shards reassemble to the full weights."""
    from tensorlink_amd.models import build_full_model, get_config
    from tensorlink_amd.models.loader import init_random_stage
    from tensorlink_amd.parallel.tp import local_config, shard_state
    cfg = get_config("tiny")
    m = build_full_model(cfg)
    init_random_stage(m, dtype=torch.float32, seed=3)
    full = m.state_dict()
    s0 = shard_state(full, cfg, 0, 2)
    s1 = shard_state(full, cfg, 1, 2)
    lc = local_config(cfg, 2)
    assert lc.num_attention_heads == 2 and lc.intermediate_size == 256
    name = "layers.0.self_attn.qkv_proj.weight"
    q, kv = cfg.q_size, cfg.kv_size
    # reassemble q/k/v blocks
    re_q = torch.cat([s0[name][:q // 2], s1[name][:q // 2]], 0)
    torch.testing.assert_close(re_q, full[name][:q])
    re_k = torch.cat([s0[name][q // 2:q // 2 + kv // 2],
                      s1[name][q // 2:q // 2 + kv // 2]], 0)
    torch.testing.assert_close(re_k, full[name][q:q + kv])
    name = "layers.0.mlp.down_proj.weight"
    re_d = torch.cat([s0[name], s1[name]], 1)
    torch.testing.assert_close(re_d, full[name])
    # replicated tensors untouched
    torch.testing.assert_close(s0["norm"], full["norm"])


def test_tunableop_noop_on_cpu():
    from tensorlink_amd.utils.tunable import setup_tunableop
    assert setup_tunableop() is False or torch.cuda.is_available()


def test_tracer_spans_and_export(tmp_path):
    """Chrome-trace export: spans, instants, gpu-arg plumbing, summary."""
    import json
    import time

    from tensorlink_amd.utils.tracing import Tracer
    tr = Tracer(rank=3, use_gpu_events=False)
    with tr.span("prefill", batch=4):
        time.sleep(0.01)
    with tr.span("decode", tokens=8):
        pass
    tr.instant("token_emitted", t=0)
    path = tr.export(str(tmp_path / "t.json"))
    data = json.load(open(path))
    names = [e["name"] for e in data["traceEvents"]]
    assert names == ["prefill", "decode", "token_emitted"]
    assert all(e["pid"] == 3 for e in data["traceEvents"])
    pre = data["traceEvents"][0]
    assert pre["ph"] == "X" and pre["dur"] >= 9_000   # >= 9 ms in us
    assert pre["args"]["batch"] == 4
    s = tr.summary()
    assert s["prefill"]["count"] == 1 and s["prefill"]["mean_ms"] >= 9


def test_runner_trace_env(tmp_path, monkeypatch):
    """TL_TRACE wires a tracer into PipelineRunner: generate() emits
    prefill + decode spans and export writes the per-rank file."""
    import json

    import torch

    from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams
    from tensorlink_amd.parallel.planner import plan_for_world
    from tensorlink_amd.utils.tracing import export_from_env
    monkeypatch.setenv("TL_TRACE", str(tmp_path / "tr"))
    r = PipelineRunner(plan_for_world("tiny", 1), 0, 1,
                       device=torch.device("cpu"))
    torch.manual_seed(1)
    r.generate(torch.randint(0, 1024, (2, 8)),
               SamplingParams(max_new_tokens=4))
    path = export_from_env(r.tracer)
    data = json.load(open(path))
    names = [e["name"] for e in data["traceEvents"]]
    assert "prefill" in names and "decode" in names


def test_auto_loaded_model_management(monkeypatch):
    """Demand-driven auto load/unload (reference
    _manage_auto_loaded_models, ml/validator.py:278): defaults load on
    the sweep, cold auto-loads unload, hot ones and explicit loads
    stay."""
    import torch

    import tensorlink_amd.config as cfg
    from tensorlink_amd.engine.engine import InferenceEngine
    monkeypatch.setattr(cfg, "DEFAULT_MODELS", ["tiny"])
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.load_model("tiny-moe")                       # explicit
    eng.manage_auto_loaded_models()
    assert "tiny" in eng.jobs                        # auto-loaded

    # demand arrives -> stays on next sweep
    eng.generate({"hf_name": "tiny", "message": "x", "max_new_tokens": 2,
                  "do_sample": False, "output_format": "simple"})
    eng.manage_auto_loaded_models()
    assert "tiny" in eng.jobs

    # no demand since the mark -> unloaded; explicit model untouched
    eng.manage_auto_loaded_models()
    assert "tiny" not in eng.jobs
    assert "tiny-moe" in eng.jobs
    eng.unload_model("tiny-moe")


def test_engine_num_beams_request():
    """num_beams>1 routes through beam search end-to-end (serial path)."""
    import torch

    from tensorlink_amd.engine.engine import InferenceEngine
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.load_model("tiny")
    out = eng.generate({"hf_name": "tiny", "message": "hello world",
                        "max_new_tokens": 5, "do_sample": False,
                        "num_beams": 3, "output_format": "simple"})
    assert "response" in out and "error" not in out
    eng.unload_model("tiny")


def test_logging_and_dashboard(tmp_path, capsys):
    """Rotating-file logger + colored console (reference debug_print,
    smart_node.py:28-125) and the status dashboard (print_ui_status,
    torch_node.py:963)."""
    import os

    import torch

    from tensorlink_amd.utils.logging import (VERBOSE, enable_file_logging,
                                              get_logger)
    lg = get_logger("tensorlink_amd.test")
    path = enable_file_logging(log_dir=str(tmp_path / "logs"),
                               name="tensorlink_amd.test")
    lg.info("hello %s", "world")
    lg.verbose("fine detail")
    lg.setLevel(VERBOSE)
    lg.verbose("now visible")
    for h in lg.handlers:
        h.flush()
    assert os.path.exists(path)
    content = open(path).read()
    assert "hello world" in content and "now visible" in content
    assert "fine detail" not in content       # below level at emit time

    from tensorlink_amd.engine.engine import InferenceEngine
    from tensorlink_amd.utils.dashboard import render_status
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.load_model("tiny")
    eng.generate({"hf_name": "tiny", "message": "x", "max_new_tokens": 2,
                  "do_sample": False, "output_format": "simple"})
    txt = render_status(eng)
    assert "tiny" in txt and "requests: 1" in txt and "[" in txt
    eng.unload_model("tiny")


def test_engine_speculative_flag():
    import torch

    from tensorlink_amd.engine.engine import InferenceEngine
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.load_model("tiny")
    a = eng.generate({"hf_name": "tiny", "message": "hello",
                      "max_new_tokens": 6, "do_sample": False,
                      "output_format": "simple"})
    b = eng.generate({"hf_name": "tiny", "message": "hello",
                      "max_new_tokens": 6, "do_sample": False,
                      "speculative": True, "output_format": "simple"})
    assert a["response"] == b["response"]
    eng.unload_model("tiny")


def test_watchdog_job_ttl_eviction():
    """Idle jobs past the TTL are unloaded on a watchdog tick (reference
    FREE_JOB_MAX_TIME job cap, validator_thread.py:19); active jobs
    survive."""
    import time

    import torch

    from tensorlink_amd.engine.engine import InferenceEngine
    from tensorlink_amd.utils.watchdog import Watchdog
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.load_model("tiny")
    eng.load_model("tiny-moe")
    time.sleep(0.15)
    # touch tiny so it is recently active
    eng.generate({"hf_name": "tiny", "message": "x", "max_new_tokens": 2,
                  "do_sample": False, "output_format": "simple"})
    wd = Watchdog(eng, job_ttl_s=0.1)
    wd.check_once()
    assert "tiny" in eng.jobs
    assert "tiny-moe" not in eng.jobs
    eng.unload_model("tiny")


def test_bench_serving_script_smoke():
    """The serving load benchmark runs end-to-end on CPU and prints a
    valid JSON metrics line."""
    import json
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "scripts/bench_serving.py", "--model", "tiny",
         "--rate", "200", "--num-requests", "6", "--prompt-len", "64",
         "--new-tokens", "4", "--prefill-chunk", "32",
         "--prefix-caching", "--shared-prefix", "40"],
        capture_output=True, text=True, timeout=240,
        cwd=__import__("os").path.dirname(__import__("os").path.dirname(
            __import__("os").path.abspath(__file__))))
    line = out.stdout.strip().splitlines()[-1]
    data = json.loads(line)
    assert data["num_requests"] == 6
    assert data["output_tokens_per_s"] > 0
    assert data["ttft_p50_s"] is not None


def test_n_completions_serial_path():
    """n>1 on the serial (non-batcher) path also yields n choices."""
    import torch

    from tensorlink_amd.engine.engine import InferenceEngine
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.load_model("tiny")
    out = eng.generate({"hf_name": "tiny", "message": "x",
                        "max_new_tokens": 4, "do_sample": True,
                        "temperature": 1.0, "n": 2,
                        "output_format": "openai"})
    assert len(out["choices"]) == 2
    eng.unload_model("tiny")


def test_batcher_tracing(tmp_path, monkeypatch):
    """TL_TRACE captures scheduler spans (prefill chunks, decode steps)
    and exports on stop."""
    import json

    import torch

    from tensorlink_amd.engine.batcher import ContinuousBatcher
    from tensorlink_amd.parallel.pipeline import PipelineRunner
    from tensorlink_amd.parallel.planner import plan_for_world
    monkeypatch.setenv("TL_TRACE", str(tmp_path / "bt"))
    r = PipelineRunner(plan_for_world("tiny", 1), 0, 1,
                       device=torch.device("cpu"))
    b = ContinuousBatcher(r, max_slots=2, max_ctx=256,
                          prefill_chunk=16).start()
    torch.manual_seed(2)
    b.submit(torch.randint(0, 1024, (40,)), max_new_tokens=4
             ).result(timeout=60)
    b.stop()
    data = json.load(open(str(tmp_path / "bt") + "_rank0.json"))
    names = {e["name"] for e in data["traceEvents"]}
    assert "prefill_chunk" in names and "decode_step" in names


def test_engine_config_serving_defaults(tmp_path):
    import json

    from tensorlink_amd.config import EngineConfig
    p = tmp_path / "config.json"
    p.write_text(json.dumps({
        "node": {"endpoint_port": 9100},
        "ml": {"continuous": True, "prefix_caching": True,
               "prefill_chunk": 256, "job_ttl_s": 60.0,
               "trusted": True}}))
    cfg = EngineConfig.load(str(p))
    assert cfg.node.endpoint_port == 9100
    assert cfg.ml.continuous and cfg.ml.prefix_caching
    assert cfg.ml.prefill_chunk == 256 and cfg.ml.job_ttl_s == 60.0
    assert cfg.ml.trusted


def test_fp8_moe_through_engine():
    """BASELINE config #5 shape on CPU: fp8-expert MoE served through
    the engine (continuous batching on) and the Qwen3-MoE family too."""
    import torch

    from tensorlink_amd.engine.engine import InferenceEngine
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.load_model("tiny-moe", quantize="fp8", continuous=True,
                   max_slots=2, max_ctx=256)
    out = eng.generate({"hf_name": "tiny-moe", "message": "hello",
                        "max_new_tokens": 4, "do_sample": False,
                        "output_format": "simple"})
    assert "response" in out and "error" not in out
    eng.unload_model("tiny-moe")
    eng.load_model("tiny-qwen3-moe", quantize="fp8")
    out = eng.generate({"hf_name": "tiny-qwen3-moe", "message": "hi",
                        "max_new_tokens": 4, "do_sample": False,
                        "output_format": "simple"})
    assert "response" in out and "error" not in out
    eng.unload_model("tiny-qwen3-moe")


def test_serve_checkpoint_directory(tmp_path):
    """load_model on a checkpoint DIRECTORY serves its weights (not a
    fresh random init): outputs match the exported source model."""
    import torch

    from tensorlink_amd.engine.engine import InferenceEngine
    from tensorlink_amd.models import build_full_model, get_config
    from tensorlink_amd.models.loader import (init_random_stage,
                                              save_hf_checkpoint)
    from tensorlink_amd.parallel.pipeline import SamplingParams

    src = build_full_model(get_config("tiny"))
    init_random_stage(src, dtype=torch.float32, seed=77)
    d = str(tmp_path / "m")
    save_hf_checkpoint(src, d)

    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    job = eng.load_model(d)
    torch.manual_seed(5)
    ids = torch.randint(0, 1024, (1, 8))
    out = job.runner.generate(ids, SamplingParams(max_new_tokens=4))
    cur = ids
    for _ in range(4):
        pos = torch.arange(cur.shape[1]).unsqueeze(0).contiguous()
        lg = src(cur, pos)
        cur = torch.cat([cur, lg[:, -1].argmax(-1, keepdim=True)], 1)
    assert torch.equal(cur[:, 8:], out)
    eng.unload_model(d)


def test_engine_two_models_concurrently():
    """Two hosted jobs serve interleaved requests (the reference keeps a
    job table of multiple hosted models — validator job lifecycle)."""
    import torch

    from tensorlink_amd.engine.engine import InferenceEngine
    eng = InferenceEngine(device=torch.device("cpu"))
    eng.load_model("tiny")
    eng.load_model("tiny-neox")
    try:
        torch.manual_seed(9)
        for name in ("tiny", "tiny-neox", "tiny", "tiny-neox"):
            out = eng.generate({"hf_name": name, "message": "ab",
                                "max_new_tokens": 4,
                                "output_format": "simple"})
            assert isinstance(out.get("response", out.get("text", "")),
                              str) or out
        assert set(eng.jobs) >= {"tiny", "tiny-neox"}
    finally:
        eng.unload_model("tiny")
        eng.unload_model("tiny-neox")
        eng.shutdown()
