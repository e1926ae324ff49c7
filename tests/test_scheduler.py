"""Scheduler math + t_index sub-sampling contract (reference lib/wrapper.py:394-407)."""
import torch

from ai_rtc_agent_amd.engine.scheduler import StreamScheduler


def test_timetable_shape_and_order():
    s = StreamScheduler(num_inference_steps=50)
    assert len(s.timesteps) == 50
    ts = s.timesteps.tolist()
    assert ts == sorted(ts, reverse=True)
    assert ts[0] == 999 and ts[-1] == 19


def test_sub_timesteps_indexing():
    s = StreamScheduler(num_inference_steps=50)
    subs = s.sub_timesteps([18, 26, 35, 45])
    assert subs == [int(s.timesteps[i]) for i in [18, 26, 35, 45]]
    # monotone decreasing in noise level (later index -> smaller timestep)
    assert subs == sorted(subs, reverse=True)


def test_repeat_interleave_law():
    # reference lib/wrapper.py:398-407: coefficients repeat per frame_buffer
    s = StreamScheduler(num_inference_steps=50)
    c1 = s.coefficients([10, 20], frame_buffer_size=1)
    c2 = s.coefficients([10, 20], frame_buffer_size=2)
    assert c2["alpha_prod_t_sqrt"].shape[0] == 4
    assert torch.equal(
        c2["alpha_prod_t_sqrt"][0], c2["alpha_prod_t_sqrt"][1]
    )
    assert torch.equal(c2["alpha_prod_t_sqrt"][0], c1["alpha_prod_t_sqrt"][0])


def test_add_noise_pred_x0_roundtrip():
    s = StreamScheduler()
    co = s.coefficients([5, 25, 45], 1)
    x0 = torch.randn(3, 8, 8, 4)
    eps = torch.randn(3, 8, 8, 4)
    x_t = s.add_noise(x0, eps, co["alpha_prod_t_sqrt"], co["beta_prod_t_sqrt"])
    x0_hat = s.pred_x0(x_t, eps, co["alpha_prod_t_sqrt"], co["beta_prod_t_sqrt"])
    assert torch.allclose(x0, x0_hat, atol=1e-5)


def test_alpha_beta_unit_energy():
    s = StreamScheduler()
    co = s.coefficients(list(range(0, 50, 7)), 1)
    e = co["alpha_prod_t_sqrt"] ** 2 + co["beta_prod_t_sqrt"] ** 2
    assert torch.allclose(e, torch.ones_like(e), atol=1e-5)


def test_cskip_cout_limits():
    s = StreamScheduler()
    co = s.coefficients([0, 49], 1)
    # high-noise stage (t≈999): c_skip ~ 0, c_out ~ 1 (prediction dominates)
    assert co["c_skip"][0] < 0.01
    assert co["c_out"][0] > 0.99
    # low-noise stage keeps more of x_t than the high-noise stage
    assert co["c_skip"][1] > co["c_skip"][0]
