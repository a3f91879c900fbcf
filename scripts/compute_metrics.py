"""Image-quality metrics between two result folders (parity with the
reference's compute_metrics.py: PSNR / LPIPS / FID).

PSNR and SSIM are computed natively (numpy/scipy). LPIPS and FID require
pretrained feature networks (AlexNet/InceptionV3) that cannot be downloaded
in this offline environment — pass --lpips_weights / --fid_weights pointing
at local TorchScript modules to enable them, otherwise they are skipped
with a note. The metric MATH is native: LPIPS = the provided module applied
to [-1,1] image pairs (reference used torchmetrics' LPIPS the same way,
/root/reference/scripts/compute_metrics.py:42-79); FID = Frechet distance
between feature means/covariances (scipy sqrtm), features from the provided
module applied to [0,1] images. Folders may contain .npy (HWC uint8) or
.png files."""

import argparse
import os

import numpy as np


def load_images(folder):
    files = sorted(
        f for f in os.listdir(folder) if f.endswith((".npy", ".png", ".jpg"))
    )
    out = {}
    for f in files:
        p = os.path.join(folder, f)
        key = os.path.splitext(f)[0]
        if key in out:
            continue
        if f.endswith(".npy"):
            out[key] = np.load(p)
        else:
            try:
                from PIL import Image

                out[key] = np.asarray(Image.open(p).convert("RGB"))
            except ImportError:
                continue
    return out


def psnr(a, b):
    mse = np.mean((a.astype(np.float64) - b.astype(np.float64)) ** 2)
    if mse == 0:
        return float("inf")
    return 10 * np.log10(255.0**2 / mse)


def ssim(a, b):
    """Global (non-windowed) SSIM — a cheap structural-similarity proxy."""
    a = a.astype(np.float64)
    b = b.astype(np.float64)
    mu_a, mu_b = a.mean(), b.mean()
    va, vb = a.var(), b.var()
    cov = ((a - mu_a) * (b - mu_b)).mean()
    c1, c2 = (0.01 * 255) ** 2, (0.03 * 255) ** 2
    return ((2 * mu_a * mu_b + c1) * (2 * cov + c2)) / (
        (mu_a**2 + mu_b**2 + c1) * (va + vb + c2)
    )


def _to_batch(imgs, keys):
    import torch

    arrs = [imgs[k].astype(np.float32) for k in keys]
    x = torch.from_numpy(np.stack(arrs)).permute(0, 3, 1, 2) / 255.0
    return x


def compute_lpips(model_path, imgs0, imgs1, keys, device="cpu"):
    """Mean LPIPS over pairs; model = TorchScript taking two [-1,1] NCHW
    batches and returning per-pair distances (or a single scalar)."""
    import torch

    model = torch.jit.load(model_path, map_location=device).eval()
    vals = []
    with torch.no_grad():
        for k in keys:
            a = _to_batch(imgs0, [k]).to(device) * 2 - 1
            b = _to_batch(imgs1, [k]).to(device) * 2 - 1
            d = model(a, b)
            vals.append(float(torch.as_tensor(d).reshape(-1).mean()))
    return float(np.mean(vals))


def frechet_distance(mu1, cov1, mu2, cov2):
    from scipy import linalg

    diff = mu1 - mu2
    covmean, _ = linalg.sqrtm(cov1 @ cov2, disp=False)
    if np.iscomplexobj(covmean):
        covmean = covmean.real
    return float(diff @ diff + np.trace(cov1) + np.trace(cov2) - 2 * np.trace(covmean))


def compute_fid(model_path, imgs0, imgs1, keys, device="cpu"):
    """FID between the two folders; model = TorchScript feature extractor
    taking a [0,1] NCHW batch and returning [N, D] features."""
    import torch

    model = torch.jit.load(model_path, map_location=device).eval()

    def feats(imgs):
        out = []
        with torch.no_grad():
            for k in keys:
                f = model(_to_batch(imgs, [k]).to(device))
                out.append(torch.as_tensor(f).reshape(1, -1).cpu().numpy())
        return np.concatenate(out, axis=0).astype(np.float64)

    f0, f1 = feats(imgs0), feats(imgs1)
    mu0, mu1 = f0.mean(0), f1.mean(0)
    cov0 = np.cov(f0, rowvar=False)
    cov1 = np.cov(f1, rowvar=False)
    cov0 = np.atleast_2d(cov0)
    cov1 = np.atleast_2d(cov1)
    return frechet_distance(mu0, cov0, mu1, cov1)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--input_root0", type=str, required=True)
    ap.add_argument("--input_root1", type=str, required=True)
    ap.add_argument("--is_gt", action="store_true", help="resize root0 to root1's size")
    ap.add_argument("--lpips_weights", type=str, default=None)
    ap.add_argument("--fid_weights", type=str, default=None)
    args = ap.parse_args()

    imgs0 = load_images(args.input_root0)
    imgs1 = load_images(args.input_root1)
    keys = sorted(set(imgs0) & set(imgs1))
    if not keys:
        raise SystemExit("no overlapping images between the two folders")

    psnrs, ssims = [], []
    for k in keys:
        a, b = imgs0[k], imgs1[k]
        if a.shape != b.shape:
            if not args.is_gt:
                raise SystemExit(f"shape mismatch for {k}: {a.shape} vs {b.shape}")
            from scipy.ndimage import zoom

            factors = [bs / as_ for bs, as_ in zip(b.shape, a.shape)]
            a = zoom(a, factors, order=1).astype(b.dtype)
        psnrs.append(psnr(a, b))
        ssims.append(ssim(a, b))

    print(f"images compared: {len(keys)}")
    print(f"PSNR: {np.mean(psnrs):.3f} dB")
    print(f"SSIM: {np.mean(ssims):.4f}")
    if args.lpips_weights is None:
        print("LPIPS: skipped (no pretrained AlexNet available offline; "
              "pass --lpips_weights <torchscript>)")
    else:
        print(f"LPIPS: {compute_lpips(args.lpips_weights, imgs0, imgs1, keys):.4f}")
    if args.fid_weights is None:
        print("FID: skipped (no pretrained InceptionV3 available offline; "
              "pass --fid_weights <torchscript>)")
    else:
        print(f"FID: {compute_fid(args.fid_weights, imgs0, imgs1, keys):.3f}")


if __name__ == "__main__":
    main()
