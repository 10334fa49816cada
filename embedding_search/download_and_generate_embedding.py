#!/usr/bin/env python3
"""Embedding-generation CLI — parity with
/root/reference/embedding_search/download_and_generate_embedding.py.

The reference downloads a LAION parquet chunk via img2dataset into
webdataset tars, embeds with the SSCD torchscript model, and dumps
`embedding.pkl` = {'features': np.float32 [N,D], 'indexes': list[str]}.
No network here: sources are a local image folder (--image_folder) or a
synthetic LAION-shaped index (--synthetic_n), same pickle contract.

Reference flag spellings (--image-folder, --dump-path, --pt-style,
--batch-size, --workers, ...) are accepted as aliases; download-phase
flags (--skip-download, --data-dir, --wandb, ...) are accepted and
ignored since there is no network / img2dataset phase here.
"""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from dcr_amd.search import generate_embeddings


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--parquet_path", "--parquet-fname", dest="parquet_path",
                   type=str, default=None,
                   help="(reference flag; requires network — unused here)")
    p.add_argument("--tars", nargs="*", default=None,
                   help="(reference flag; webdataset tars — unused here)")
    p.add_argument("--image_folder", "--image-folder", dest="image_folder",
                   type=str, default=None)
    p.add_argument("--synthetic_n", type=int, default=None)
    p.add_argument("--dump_path", "--dump-path", dest="dump_path",
                   type=str, required=True)
    p.add_argument("--pt_model", "--pt-style", dest="pt_model",
                   type=str, default="sscd",
                   choices=["sscd", "sscd_im", "sscd_disc_large"])
    p.add_argument("--arch", type=str, default="resnet50",
                   choices=["resnet50"],
                   help="(reference flag; SSCD backbone arch)")
    p.add_argument("--batch_size", "--batch-size", dest="batch_size",
                   type=int, default=128)
    p.add_argument("--num_workers", "--workers", dest="num_workers",
                   type=int, default=4)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--gpu", type=int, default=0,
                   help="device index (reference flag)")
    p.add_argument("--delete_tars", action="store_true")
    # accepted-and-ignored reference flags (download/wandb phases don't
    # exist here; reference's own --multiscale embedding path is broken —
    # SURVEY.md §2.6.5)
    for flag in ("--wandb", "--skip-download", "--skip-img-embed",
                 "--skip-image-delete", "--multiscale"):
        p.add_argument(flag, action="store_true",
                       help="(reference flag; no-op here)")
    p.add_argument("--similarity-metric", type=str, default="d",
                   help="(reference flag; no-op here)")
    p.add_argument("--data-dir", type=str, default=None,
                   help="(reference flag; no-op here)")
    args = p.parse_args()
    if args.image_folder is None and args.synthetic_n is None:
        raise SystemExit("need --image_folder or --synthetic_n (no network for "
                         "the parquet/img2dataset path)")
    import torch
    if torch.cuda.is_available() and args.gpu:
        torch.cuda.set_device(args.gpu)
    blob = generate_embeddings(
        args.image_folder, Path(args.dump_path) / "embedding.pkl",
        pt_model=args.pt_model, batch_size=args.batch_size,
        num_workers=args.num_workers, synthetic_n=args.synthetic_n,
        seed=args.seed)
    print(f"embedded {blob['features'].shape} -> {args.dump_path}/embedding.pkl")


if __name__ == "__main__":
    main()
