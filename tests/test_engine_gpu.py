"""GPU integration tests: the engine through the HIP kernels + hipGraph.

SURVEY.md §4 item (c): single-GPU integration on synthetic video with
random-init weights.
"""
import pytest
import torch

from ai_rtc_agent_amd.config import EngineConfig, sd_turbo_config
from ai_rtc_agent_amd.engine import StreamDiffusionEngine
from ai_rtc_agent_amd.models import UNet2DCondition, UNetConfig

pytestmark = pytest.mark.gpu


def frame(h=512, w=512, seed=0, device="cuda"):
    g = torch.Generator().manual_seed(seed)
    return torch.randint(0, 256, (h, w, 3), generator=g, dtype=torch.uint8).to(device)


def test_sd_turbo_end_to_end():
    cfg = sd_turbo_config(device="cuda")
    e = StreamDiffusionEngine(cfg)
    e.prepare()
    out = e(frame(seed=1))
    assert out.shape == (512, 512, 3) and out.dtype == torch.uint8
    assert out.is_cuda  # zero-copy contract: output stays in HBM (-> encoder)


def test_graph_replay_matches_eager():
    """hipGraph capture (pipelined two-stream mode) must be bit-identical to
    the same kernels eager. Outputs are cloned after sync_output: pipelined
    outputs live in ping-pong buffers valid until the same-parity frame two
    calls later."""
    outs = {}
    for use_graph in (False, True):
        cfg = sd_turbo_config(device="cuda", use_hip_graph=use_graph)
        e = StreamDiffusionEngine(cfg)
        e.prepare()
        res = []
        for i in range(4):
            out = e(frame(seed=10 + i))
            e.sync_output()
            res.append(out.clone())
        outs[use_graph] = res
    for i, (a, b) in enumerate(zip(outs[False], outs[True])):
        assert torch.equal(a, b), f"graph replay diverged from eager at frame {i}"


def test_pipelined_vs_sequential_graph():
    """Two-stream pipelined replay must produce the same frames as the
    single-graph sequential replay."""
    outs = {}
    for overlap in (False, True):
        cfg = sd_turbo_config(device="cuda", use_hip_graph=True)
        cfg.pipeline_overlap = overlap
        e = StreamDiffusionEngine(cfg)
        e.prepare()
        res = []
        for i in range(6):
            out = e(frame(seed=40 + i))
            e.sync_output()
            res.append(out.clone())
        outs[overlap] = res
    for i, (a, b) in enumerate(zip(outs[False], outs[True])):
        assert torch.equal(a, b), f"pipelined diverged at frame {i}"


def _run_clone(e, f, n=3):
    out = None
    for _ in range(n):
        out = e(f)
    e.sync_output()
    return out.clone()


def test_prompt_update_through_graph():
    cfg = sd_turbo_config(device="cuda")
    e = StreamDiffusionEngine(cfg)
    e.prepare()
    f = frame(seed=3)
    base = _run_clone(e, f)
    e.update_prompt("an entirely different style prompt")
    after = _run_clone(e, f)
    assert not torch.equal(base, after), "graph-external embed update must take effect"


def test_sd15_4step_lcm_config():
    """BASELINE config[2]-shaped run (SD1.5 4-step + similarity filter)."""
    cfg = EngineConfig(device="cuda", model_family="sd15", use_lcm_lora=True)
    cfg.similarity_filter.enabled = True
    e = StreamDiffusionEngine(cfg)
    e.prepare()
    f = frame(seed=5)
    for i in range(3):
        out = e(f)
    assert out.shape == (512, 512, 3)


def test_unet_gpu_matches_cpu_golden():
    """fp16 HIP UNet vs fp32 torch reference (tiny config, loose tolerance)."""
    torch.manual_seed(0)
    cfg = UNetConfig.tiny()
    net = UNet2DCondition(cfg).eval()
    x = torch.randn(1, 16, 16, 4)
    t = torch.tensor([500])
    ctx = torch.randn(1, 77, cfg.cross_attention_dim)
    with torch.no_grad():
        ref = net(x, t, ctx)
        gpu = net.to("cuda").half()
        got = gpu(x.cuda().half(), t.cuda(), ctx.cuda().half()).float().cpu()
    corr = torch.corrcoef(torch.stack([ref.flatten(), got.flatten()]))[0, 1]
    assert corr > 0.99, f"UNet GPU/CPU correlation {corr}"
    assert (ref - got).abs().mean() < 0.05


def test_txt2img_gpu():
    cfg = sd_turbo_config(device="cuda")
    cfg.mode = "txt2img"
    e = StreamDiffusionEngine(cfg)
    e.prepare()
    out = e.txt2img()
    assert out.shape == (1, 512, 512, 3)


def test_agent_serving_path_on_gpu():
    """The serving stack (PipelinePool -> StreamDiffusionPipeline -> engine)
    on a real GPU: what the agent runs per frame."""
    from ai_rtc_agent_amd.config import sd_turbo_config
    from ai_rtc_agent_amd.parallel.dispatch import PipelinePool

    cfg = sd_turbo_config(device="cuda")
    pool = PipelinePool.create(model_id="stabilityai/sd-turbo", n_gpus=1, cfg=cfg)
    p = pool.assign("stream-1")
    f = frame(seed=77)
    out = p(f)
    assert out.shape == (512, 512, 3) and out.is_cuda
    stats = pool.stats()
    assert stats["replicas"] == 1
    assert stats["per_replica"][0]["frames"] >= 1
    pool.release("stream-1")


def test_t_index_update_and_recapture_on_gpu():
    """Runtime t_index updates under graph mode: same-length updates write
    coefficient buffers in place (no re-capture); a length change triggers
    a clean re-prepare + re-capture (reference lib/wrapper.py:389-407)."""
    cfg = EngineConfig(
        device="cuda", model_family="sd21", model_id="stabilityai/sd-turbo",
        t_index_list=[10, 30], num_inference_steps=50, cfg_type="none",
        use_lcm_lora=False,
    )
    e = StreamDiffusionEngine(cfg)
    e.prepare()
    f = frame(seed=9)
    base = _run_clone(e, f)
    e.update_t_index_list([5, 45])  # same length: in-place, graph kept
    after = _run_clone(e, f)
    assert not torch.equal(base, after)
    e.update_t_index_list([0])      # length change: full re-prepare
    out = _run_clone(e, f)
    assert out.shape == (512, 512, 3)


def test_lora_hot_swap_recaptures_graph():
    """Weight hot-swap under graph mode: caches invalidated, graphs
    re-captured, outputs change; the engine keeps serving."""
    from ai_rtc_agent_amd.models.lora import make_random_lora

    cfg = sd_turbo_config(device="cuda")
    e = StreamDiffusionEngine(cfg)
    e.prepare()
    f = frame(seed=21)
    base = _run_clone(e, f)
    sd = make_random_lora(e.unet, rank=4, seed=5, limit=30)
    assert e.load_lora(sd, scale=3.0) > 0
    after = _run_clone(e, f)
    assert not torch.equal(base, after)
    # still bit-stable frame to frame after the re-capture
    again = _run_clone(e, f)
    assert again.shape == (512, 512, 3)


@pytest.mark.timeout(280)
def test_rccl_two_ranks():
    """REAL RCCL on hardware: two torchrun ranks broadcast engine weights
    over RCCL and all-reduce the bench timings — the collective path the
    8-GPU scale-out uses. RCCL requires one DISTINCT device per
    communicator rank (two ranks on one GPU is "invalid usage" by
    design), so this runs only on multi-GPU boxes; the single-GPU pool
    covers the logic with 2-rank gloo tests instead."""
    import json
    import os
    import socket
    import subprocess
    import sys

    import torch

    if torch.cuda.device_count() < 2:
        pytest.skip("RCCL needs a distinct device per rank")

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    env = dict(os.environ, AIRTC_BENCH_FAMILY="tiny",
               MASTER_ADDR="127.0.0.1")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), "bench.py", "--gpus", "2",
         "--steps", "6", "--warmup", "2", "--latency-frames", "2",
         "--e2e-frames", "0", "--width", "64"],
        capture_output=True, text=True, timeout=240,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        env=env)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2
    assert d["value"] > 0


def test_fp8_tier_end_to_end_on_gpu():
    """fp8 serving tier on real hardware: calibrate on the first frames
    (eager), freeze + quality gate, capture the GN-fp8 -> MX-MFMA conv
    graph, and keep serving. Output must track the f16 engine closely
    (same seed/weights; fp8 noise only)."""
    cfg16 = sd_turbo_config(device="cuda")
    e16 = StreamDiffusionEngine(cfg16)
    e16.prepare()
    cfg8 = sd_turbo_config(device="cuda", use_fp8=True, fp8_calib_frames=2)
    e8 = StreamDiffusionEngine(cfg8)
    e8.prepare()
    outs16, outs8 = [], []
    for i in range(5):
        f = frame(seed=40 + i)
        o16 = e16(f)
        e16.sync_output()
        outs16.append(o16.float().clone())
        o8 = e8(f)
        e8.sync_output()
        outs8.append(o8.float().clone())
    assert e8.fp8_active, f"gate failed: {e8.fp8_snr_db} dB"
    assert e8.fp8_snr_db > cfg8.fp8_min_snr_db
    # after capture (frame >= calib+1) the fp8 graph serves; outputs are
    # u8 images whose difference from f16 is bounded quantization noise
    d = (outs8[-1] - outs16[-1]).abs().mean().item()
    assert d < 24.0, f"fp8 output drifted {d} u8 steps from f16"
    st = e8.stats()
    assert st["fp8"]["active"] and not st["fp8"]["calibrating"]
