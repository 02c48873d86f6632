"""Vision transform ops (tensor-native re-expression of the reference's
preprocess.py set) + config-driven composition."""

import numpy as np
import torch

from paddlefleetx_amd.data.transforms import (CenterCropImage, ColorJitter,
                                              Compose, DecodeImage,
                                              GaussianBlur, NormalizeImage,
                                              RandCropImage, RandFlipImage,
                                              ResizeImage, ToCHWImage,
                                              build_transforms)


def test_decode_resize_crop_shapes():
    raw = (np.random.rand(40, 60, 3) * 255).astype(np.uint8)  # HWC uint8
    img = DecodeImage()(raw)
    assert img.shape == (3, 40, 60) and img.max() <= 1.0
    r = ResizeImage(resize_short=32)(img)
    assert min(r.shape[1:]) == 32
    c = CenterCropImage(24)(r)
    assert c.shape == (3, 24, 24)
    s = ResizeImage(size=16)(img)
    assert s.shape == (3, 16, 16)


def test_rand_crop_and_flip_deterministic():
    g = torch.Generator().manual_seed(0)
    img = torch.rand(3, 48, 48)
    rc = RandCropImage(size=32, generator=g)
    out = rc(img)
    assert out.shape == (3, 32, 32)
    g1 = torch.Generator().manual_seed(1)
    g2 = torch.Generator().manual_seed(1)
    f1 = RandFlipImage(generator=g1)(img)
    f2 = RandFlipImage(generator=g2)(img)
    assert torch.equal(f1, f2)  # same seed -> same flip decision


def test_normalize_and_scale_string():
    img = torch.full((3, 4, 4), 128.0)  # uint8-range values
    n = NormalizeImage(scale="1./255.", mean=[0.5, 0.5, 0.5],
                       std=[0.5, 0.5, 0.5])
    out = n(img)
    expected = (128.0 / 255.0 - 0.5) / 0.5
    assert torch.allclose(out, torch.full_like(out, expected), atol=1e-6)


def test_color_jitter_and_blur_bounded():
    g = torch.Generator().manual_seed(3)
    img = torch.rand(3, 16, 16)
    cj = ColorJitter(brightness=0.4, contrast=0.4, saturation=0.4, hue=0.1,
                     generator=g)
    out = cj(img)
    assert out.shape == img.shape and torch.isfinite(out).all()
    assert out.min() >= 0 and out.max() <= 1.0 + 1e-6
    gb = GaussianBlur(sigma=(0.5, 0.5), generator=g)
    blur = gb(img)
    assert blur.shape == img.shape
    # blur reduces variance but preserves the mean (up to edge effects)
    assert blur.var() < img.var()
    assert abs(float(blur.mean() - img.mean())) < 0.05


def test_build_transforms_from_config():
    pipeline = build_transforms([
        {"DecodeImage": {"to_rgb": True}},
        {"RandCropImage": {"size": 24}},
        {"RandFlipImage": {"flip_code": 1}},
        {"NormalizeImage": {"scale": "1./255.",
                            "mean": [0.485, 0.456, 0.406],
                            "std": [0.229, 0.224, 0.225]}},
    ], generator=torch.Generator().manual_seed(0))
    assert isinstance(pipeline, Compose) and len(pipeline.ops) == 4
    raw = (np.random.rand(32, 32, 3) * 255).astype(np.uint8)
    out = pipeline(raw)
    assert out.shape == (3, 24, 24)
    # normalized output is roughly centred
    assert abs(float(out.mean())) < 3.0

    chw = ToCHWImage()(raw)
    assert chw.shape == (3, 32, 32)

    try:
        build_transforms([{"NoSuchOp": {}}])
        raise AssertionError("expected ValueError")
    except ValueError:
        pass


def test_contrastive_dataset_two_views(tmp_path):
    import os
    from paddlefleetx_amd.data.vision_dataset import ContrativeLearningDataset
    os.makedirs(tmp_path / "cls0")
    np.save(tmp_path / "cls0" / "a.npy",
            np.random.rand(3, 32, 32).astype("float32"))
    ds = ContrativeLearningDataset(str(tmp_path), transform_ops=[
        {"RandCropImage": {"size": 24}},
        {"ColorJitter": {"brightness": 0.4, "p": 0.8}},
    ])
    (q, k), label = ds[0]
    assert q.shape == (3, 24, 24) and k.shape == (3, 24, 24)
    assert label == 0


def test_cifar10_dataset(tmp_path):
    import pickle
    from paddlefleetx_amd.data.vision_dataset import CIFAR10Dataset
    data = (np.random.rand(20, 3072) * 255).astype(np.uint8)
    with open(tmp_path / "data_batch_1", "wb") as f:
        pickle.dump({b"data": data, b"labels": list(range(10)) * 2}, f)
    ds = CIFAR10Dataset(str(tmp_path), mode="Train",
                        transform_ops=[{"RandFlipImage": {}}])
    assert len(ds) == 20
    img, label = ds[3]
    assert img.shape == (3, 32, 32) and 0 <= label < 10
    assert img.max() <= 1.0
