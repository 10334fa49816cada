"""Scheduler math tests (CPU): add_noise/velocity semantics
(/root/reference/diff_train.py:632,650) and sampler sanity."""
import torch

from dcr_amd.schedulers import DDIMScheduler, DDPMScheduler, DPMSolverMultistepScheduler


def test_add_noise_matches_closed_form():
    s = DDPMScheduler()
    torch.manual_seed(0)
    x0 = torch.randn(4, 4, 8, 8)
    noise = torch.randn_like(x0)
    t = torch.tensor([0, 10, 500, 999])
    xt = s.add_noise(x0, noise, t)
    ac = s.alphas_cumprod[t].view(-1, 1, 1, 1)
    ref = ac.sqrt() * x0 + (1 - ac).sqrt() * noise
    assert torch.allclose(xt, ref, atol=1e-5)


def test_velocity_matches_closed_form():
    s = DDPMScheduler()
    x0 = torch.randn(3, 4, 8, 8)
    noise = torch.randn_like(x0)
    t = torch.tensor([1, 300, 998])
    v = s.get_velocity(x0, noise, t)
    ac = s.alphas_cumprod[t].view(-1, 1, 1, 1)
    ref = ac.sqrt() * noise - (1 - ac).sqrt() * x0
    assert torch.allclose(v, ref, atol=1e-5)


def test_add_noise_t0_near_identity():
    s = DDPMScheduler()
    x0 = torch.randn(1, 4, 8, 8)
    noise = torch.randn_like(x0)
    xt = s.add_noise(x0, noise, torch.tensor([0]))
    # ac[0] = 1 - 0.00085 => sqrt(1-ac) ~ 0.029, so x_t stays close to x0
    assert (xt - x0).abs().max() < 0.029 * noise.abs().max() + 0.01


def test_ddim_deterministic_denoise_identity_model():
    """With a model predicting the exact noise, 'denoising' pure-noise-free
    data recovers x0 when starting from x_t built by add_noise."""
    s = DDIMScheduler()
    s.set_timesteps(10)
    x = torch.randn(1, 4, 8, 8)
    for t in s.timesteps:
        eps = torch.zeros_like(x)  # model says "no noise present"
        x = s.step(eps, int(t), x).prev_sample
    assert torch.isfinite(x).all()


def test_dpm_solver_runs_and_finite():
    s = DPMSolverMultistepScheduler()
    s.set_timesteps(10)
    assert len(s.timesteps) == 10
    x = torch.randn(2, 4, 8, 8)
    for t in s.timesteps:
        x = s.step(torch.randn_like(x) * 0.1, int(t), x).prev_sample
    assert torch.isfinite(x).all()


def test_scheduler_config_roundtrip(tmp_path):
    s = DDPMScheduler(prediction_type="v_prediction")
    s.save_pretrained(tmp_path / "scheduler")
    s2 = DDPMScheduler.from_pretrained(tmp_path / "scheduler")
    assert s2.prediction_type == "v_prediction"
    assert torch.allclose(s.alphas_cumprod, s2.alphas_cumprod)


def test_v_prediction_target_path():
    """reference diff_train.py:647-652: target switches with prediction_type"""
    s = DDPMScheduler(prediction_type="v_prediction")
    assert s.prediction_type == "v_prediction"


def test_ddim_eta_adds_noise():
    """eta>0 ancestral path: deterministic part identical, noise term added."""
    s = DDIMScheduler()
    s.set_timesteps(10)
    x = torch.randn(1, 4, 8, 8)
    eps = torch.randn_like(x) * 0.2
    t = int(s.timesteps[4])
    det = s.step(eps, t, x, eta=0.0).prev_sample
    g = torch.Generator().manual_seed(0)
    sto = s.step(eps, t, x, eta=1.0, generator=g).prev_sample
    assert not torch.allclose(det, sto)
    # noise magnitude bounded by sigma
    ac_t = s.alphas_cumprod[t]
    prev_t = t - s.config.num_train_timesteps // s.num_inference_steps
    ac_p = s.alphas_cumprod[prev_t]
    sigma = ((1 - ac_p) / (1 - ac_t) * (1 - ac_t / ac_p)).sqrt()
    assert (sto - det).abs().max() < 6 * sigma


def test_dpm_v_prediction_runs():
    s = DPMSolverMultistepScheduler(prediction_type="v_prediction")
    s.set_timesteps(6)
    x = torch.randn(1, 4, 8, 8)
    for t in s.timesteps:
        x = s.step(torch.randn_like(x) * 0.1, int(t), x).prev_sample
    assert torch.isfinite(x).all()


def test_ddpm_ancestral_step_t0_deterministic():
    s = DDPMScheduler()
    x = torch.randn(1, 4, 8, 8)
    eps = torch.randn_like(x)
    out1 = s.step(eps, 0, x)
    out2 = s.step(eps, 0, x)
    assert torch.equal(out1, out2)  # t=0 adds no noise


def test_ddim_full_denoise_recovers_clean_direction():
    """denoising x_t built from known x0/eps with a perfect eps-model
    moves the sample toward x0."""
    s = DDIMScheduler()
    s.set_timesteps(50)
    ddpm = DDPMScheduler()
    torch.manual_seed(0)
    x0 = torch.randn(1, 4, 8, 8)
    eps = torch.randn_like(x0)
    t0 = int(s.timesteps[0])
    x = ddpm.add_noise(x0, eps, torch.tensor([t0]))
    for t in s.timesteps:
        # oracle epsilon for the CURRENT x given fixed x0
        ac = s.alphas_cumprod[int(t)]
        eps_hat = (x - ac.sqrt() * x0) / (1 - ac).sqrt()
        x = s.step(eps_hat, int(t), x).prev_sample
    # final step keeps sqrt(1-ac[0]) ~ 0.029 of eps (set_alpha_to_one=False)
    assert (x - x0).abs().max() < 0.15, (x - x0).abs().max()
