"""Imagen: continuous-time Gaussian diffusion + text-conditioned U-Net.

Reference: ppfleetx/models/multimodal_model/imagen/modeling.py —
ImagenCriterion :94 (p2-weighted eps-MSE), ImagenModel :138 (T5/DebertaV2
text encoder + unet + GaussianDiffusionContinuousTimes from utils.py:384).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from paddlefleetx_amd.models.imagen.unet import Unet


def log_snr_cosine(t: torch.Tensor, s: float = 0.008) -> torch.Tensor:
    """Cosine noise schedule in log-SNR form (utils.py:384
    GaussianDiffusionContinuousTimes, alpha_cosine_log_snr)."""
    return -torch.log(
        torch.clamp(torch.tan((t + s) / (1 + s) * math.pi / 2) ** 2,
                    min=1e-8, max=1e8))


class GaussianDiffusionContinuousTimes(nn.Module):
    def __init__(self, noise_schedule: str = "cosine", timesteps: int = 1000):
        super().__init__()
        assert noise_schedule == "cosine"
        self.num_timesteps = timesteps

    def sample_random_times(self, batch: int, device) -> torch.Tensor:
        return torch.rand(batch, device=device)

    def log_snr(self, t):
        return log_snr_cosine(t)

    def alpha_sigma(self, t):
        log_snr = self.log_snr(t)
        alpha = torch.sqrt(torch.sigmoid(log_snr))
        sigma = torch.sqrt(torch.sigmoid(-log_snr))
        return alpha, sigma

    def q_sample(self, x0, t, noise=None):
        noise = noise if noise is not None else torch.randn_like(x0)
        alpha, sigma = self.alpha_sigma(t)
        a = alpha[:, None, None, None]
        s = sigma[:, None, None, None]
        return a * x0 + s * noise, noise

    def predict_start_from_noise(self, x_t, t, noise):
        alpha, sigma = self.alpha_sigma(t)
        a = alpha[:, None, None, None]
        s = sigma[:, None, None, None]
        return (x_t - s * noise) / a.clamp(min=1e-8)

    def q_posterior_mean(self, x0, x_t, t, t_next):
        """DDPM-style posterior mean for the t -> t_next step."""
        log_snr_t = self.log_snr(t)
        log_snr_n = self.log_snr(t_next)
        alpha_t = torch.sqrt(torch.sigmoid(log_snr_t))
        alpha_n = torch.sqrt(torch.sigmoid(log_snr_n))
        sigma_t2 = torch.sigmoid(-log_snr_t)
        sigma_n2 = torch.sigmoid(-log_snr_n)
        c = -torch.expm1(log_snr_t - log_snr_n)
        mean = alpha_n[:, None, None, None] * (
            x_t * (1 - c)[:, None, None, None] /
            alpha_t[:, None, None, None].clamp(min=1e-8)
            + c[:, None, None, None] * x0)
        var = sigma_n2 * c
        return mean, var


class ImagenCriterion(nn.Module):
    """p2-weighted eps-prediction MSE (modeling.py:94-137)."""

    def __init__(self, p2_loss_weight_gamma: float = 0.5,
                 p2_loss_weight_k: float = 1.0):
        super().__init__()
        self.gamma = p2_loss_weight_gamma
        self.k = p2_loss_weight_k

    def forward(self, pred, target, log_snr=None):
        loss = F.mse_loss(pred.float(), target.float(), reduction="none")
        loss = loss.mean(dim=tuple(range(1, loss.ndim)))
        if self.gamma > 0 and log_snr is not None:
            w = (self.k + log_snr.exp()) ** -self.gamma
            loss = loss * w
        return loss.mean()


class ImagenModel(nn.Module):
    """One Imagen stage: text encoder + denoising U-Net (modeling.py:138)."""

    def __init__(self, unet: Optional[Unet] = None, image_size: int = 64,
                 text_encoder_name: str = "t5", text_embed_dim: int = 512,
                 timesteps: int = 1000, cond_drop_prob: float = 0.1,
                 text_encoder_kwargs: Optional[dict] = None,
                 unet_kwargs: Optional[dict] = None,
                 freeze_text_encoder: bool = True,
                 dynamic_thresholding: bool = True,
                 dynamic_thresholding_percentile: float = 0.95, **unused):
        super().__init__()
        self.image_size = image_size
        self.cond_drop_prob = cond_drop_prob
        # reference modeling.py:289-292, 378-383
        self.dynamic_thresholding = dynamic_thresholding
        self.dynamic_thresholding_percentile = dynamic_thresholding_percentile
        if unet is None:
            unet = Unet(text_embed_dim=text_embed_dim, **(unet_kwargs or {}))
        self.unet = unet
        from paddlefleetx_amd.models.t5 import T5EncoderModel
        self.text_encoder = T5EncoderModel(
            d_model=text_embed_dim, **(text_encoder_kwargs or {}))
        if freeze_text_encoder:
            for p in self.text_encoder.parameters():
                p.requires_grad = False
        self.scheduler = GaussianDiffusionContinuousTimes(
            timesteps=timesteps)
        self.criterion = ImagenCriterion()

    def encode_text(self, text_ids, text_mask=None):
        with torch.no_grad():
            return self.text_encoder(text_ids, attention_mask=text_mask)

    def forward(self, images, text_ids=None, text_mask=None,
                text_embeds=None, lowres_images=None):
        """Training loss for one denoising step. SR stages
        (unet.lowres_cond) take `lowres_images` (or derive them by
        downsampling the targets) with noise augmentation — reference
        modeling.py lowres conditioning path."""
        B = images.shape[0]
        device = images.device
        if text_embeds is None and text_ids is not None:
            text_embeds = self.encode_text(text_ids, text_mask)
        if text_embeds is not None and self.training and \
                self.cond_drop_prob > 0:
            keep = (torch.rand(B, device=device) >
                    self.cond_drop_prob)[:, None, None]
            text_embeds = text_embeds * keep
        lowres_cond_img = None
        if getattr(self.unet, "lowres_cond", False):
            if lowres_images is None:
                lowres_images = F.interpolate(
                    images, scale_factor=0.25, mode="bilinear",
                    align_corners=False)
            lowres_cond_img = F.interpolate(
                lowres_images, size=images.shape[-2:], mode="bilinear",
                align_corners=False)
            if self.training:
                # lowres noise augmentation (aug time <= 0.5)
                aug_t = torch.rand(B, device=device) * 0.5
                lowres_cond_img, _ = self.scheduler.q_sample(
                    lowres_cond_img, aug_t)
        t = self.scheduler.sample_random_times(B, device)
        x_t, noise = self.scheduler.q_sample(images, t)
        pred = self.unet(x_t.to(images.dtype), t, text_embeds=text_embeds,
                         text_mask=text_mask, lowres_cond_img=lowres_cond_img)
        return self.criterion(pred, noise, log_snr=self.scheduler.log_snr(t))

    @torch.no_grad()
    def sample(self, text_ids=None, text_mask=None, batch_size: int = 1,
               steps: int = 50, device=None, lowres_images=None):
        device = device or next(self.unet.parameters()).device
        text_embeds = self.encode_text(text_ids.to(device), text_mask) \
            if text_ids is not None else None
        lowres_cond_img = None
        if getattr(self.unet, "lowres_cond", False):
            assert lowres_images is not None, \
                "SR stage sampling needs the previous stage's images"
            lowres_cond_img = F.interpolate(
                lowres_images.to(device), size=(self.image_size,
                                                self.image_size),
                mode="bilinear", align_corners=False)
            batch_size = lowres_cond_img.shape[0]
        x = torch.randn(batch_size, self.unet.channels, self.image_size,
                        self.image_size, device=device)
        times = torch.linspace(1.0, 0.0, steps + 1, device=device)
        for i in range(steps):
            t = times[i].repeat(batch_size)
            t_next = times[i + 1].repeat(batch_size)
            eps = self.unet(x, t, text_embeds=text_embeds,
                            text_mask=text_mask,
                            lowres_cond_img=lowres_cond_img)
            x0 = self.scheduler.predict_start_from_noise(x, t, eps)
            if self.dynamic_thresholding:
                # per-sample percentile of |x0|, clamp + renormalize
                # (reference modeling.py:378-383)
                s = torch.quantile(
                    x0.reshape(x0.shape[0], -1).abs().float(),
                    self.dynamic_thresholding_percentile, dim=-1)
                s = s.clamp(min=1.0)[:, None, None, None].to(x0.dtype)
                x0 = x0.clamp(-s, s) / s
            else:
                x0 = x0.clamp(-1, 1)
            mean, var = self.scheduler.q_posterior_mean(x0, x, t, t_next)
            noise = torch.randn_like(x) if i < steps - 1 else 0
            x = mean + var.sqrt()[:, None, None, None] * noise
        return x


class ImagenCascade(nn.Module):
    """Cascading DDPM (reference modeling.py:976-1035 Imagen 'base' +
    SRUnet256 [+ SRUnet1024] chains): one text encoder shared across
    stages, per-stage schedulers, chained sampling 64 -> 256 (-> 1024).
    """

    def __init__(self, unets, image_sizes, text_embed_dim: int = 512,
                 text_encoder_kwargs: Optional[dict] = None,
                 timesteps: int = 1000, **stage_kwargs):
        super().__init__()
        assert len(unets) == len(image_sizes)
        self.stages = nn.ModuleList()
        for i, (unet, size) in enumerate(zip(unets, image_sizes)):
            stage = ImagenModel(unet=unet, image_size=size,
                                text_embed_dim=text_embed_dim,
                                text_encoder_kwargs=text_encoder_kwargs,
                                timesteps=timesteps, **stage_kwargs)
            if i > 0:
                # share the (frozen) text encoder with stage 0
                stage.text_encoder = self.stages[0].text_encoder
            self.stages.append(stage)
        self.image_sizes = list(image_sizes)

    def forward(self, images, text_ids=None, text_mask=None,
                unet_number: int = 0):
        """Train one stage: the target images are resized to the stage's
        resolution; SR stages condition on the previous stage's size."""
        stage = self.stages[unet_number]
        size = self.image_sizes[unet_number]
        tgt = F.interpolate(images, size=(size, size), mode="bilinear",
                            align_corners=False) \
            if images.shape[-1] != size else images
        lowres = None
        if unet_number > 0:
            prev = self.image_sizes[unet_number - 1]
            lowres = F.interpolate(images, size=(prev, prev),
                                   mode="bilinear", align_corners=False)
        return stage(tgt, text_ids=text_ids, text_mask=text_mask,
                     lowres_images=lowres)

    @torch.no_grad()
    def sample(self, text_ids=None, text_mask=None, batch_size: int = 1,
               steps: int = 25, stop_at_unet_number: Optional[int] = None,
               device=None):
        img = None
        for i, stage in enumerate(self.stages):
            img = stage.sample(text_ids=text_ids, text_mask=text_mask,
                               batch_size=batch_size, steps=steps,
                               device=device, lowres_images=img)
            if stop_at_unet_number is not None and \
                    i + 1 >= stop_at_unet_number:
                break
        return img
