"""Dump ground-truth images + prompts.json from a LOCAL COCO-layout dataset
(parity with the reference's dump_coco.py, which streamed HuggingFaceM4/COCO —
no network here, so point --coco_root at a local directory containing
annotations/captions_val2014.json and val2014/)."""

import argparse
import json
import os
import shutil


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--coco_root", type=str, required=True)
    ap.add_argument("--output_root", type=str, default="results/coco/gt")
    ap.add_argument("--n_images", type=int, default=5000)
    args = ap.parse_args()

    ann_path = os.path.join(args.coco_root, "annotations", "captions_val2014.json")
    with open(ann_path) as f:
        ann = json.load(f)
    id_to_file = {im["id"]: im["file_name"] for im in ann["images"]}
    os.makedirs(args.output_root, exist_ok=True)
    prompts = {}
    count = 0
    for a in ann["annotations"]:
        if count >= args.n_images:
            break
        img_id = a["image_id"]
        if img_id not in id_to_file:
            continue
        src = os.path.join(args.coco_root, "val2014", id_to_file[img_id])
        if not os.path.exists(src):
            continue
        dst = os.path.join(args.output_root, f"{count:05d}{os.path.splitext(src)[1]}")
        shutil.copyfile(src, dst)
        prompts[f"{count:05d}"] = a["caption"]
        count += 1
    with open(os.path.join(args.output_root, "prompts.json"), "w") as f:
        json.dump(prompts, f, indent=1)
    print(f"dumped {count} images -> {args.output_root}")


if __name__ == "__main__":
    main()
