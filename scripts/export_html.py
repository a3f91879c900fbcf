"""Side-by-side HTML comparison table of result folders (parity with the
reference's export_html.py, plain-string templating instead of dominate)."""

import argparse
import json
import os


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--input_roots", type=str, nargs="+", required=True)
    ap.add_argument("--names", type=str, nargs="*", default=None)
    ap.add_argument("--prompt_file", type=str, default=None)
    ap.add_argument("--output_path", type=str, default="compare.html")
    ap.add_argument("--max_images", type=int, default=100)
    args = ap.parse_args()

    names = args.names or [os.path.basename(r.rstrip("/")) for r in args.input_roots]
    prompts = {}
    if args.prompt_file:
        with open(args.prompt_file) as f:
            prompts = json.load(f)

    keys = None
    for root in args.input_roots:
        files = {os.path.splitext(f)[0] for f in os.listdir(root)
                 if f.endswith((".png", ".jpg"))}
        keys = files if keys is None else keys & files
    keys = sorted(keys)[: args.max_images]

    rows = []
    for k in keys:
        cells = "".join(
            f'<td><img src="{os.path.join(root, k + ".png")}" width="256"></td>'
            for root in args.input_roots
        )
        caption = prompts.get(k, k)
        rows.append(f"<tr><td>{caption}</td>{cells}</tr>")

    header = "".join(f"<th>{n}</th>" for n in names)
    html = (
        "<html><head><style>table{border-collapse:collapse}"
        "td,th{border:1px solid #999;padding:4px;font-family:sans-serif}</style>"
        f"</head><body><table><tr><th>prompt</th>{header}</tr>"
        + "".join(rows)
        + "</table></body></html>"
    )
    with open(args.output_path, "w") as f:
        f.write(html)
    print(f"wrote {args.output_path} ({len(keys)} rows)")


if __name__ == "__main__":
    main()
