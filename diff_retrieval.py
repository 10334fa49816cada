#!/usr/bin/env python3
"""Retrieval/metrics CLI — parity with /root/reference/diff_retrieval.py.

Distributed feature extraction of generated + training images (SSCD /
DINO / CLIP backbones), similarity matrices (rocBLAS GEMM), similarity
stats + histograms, CLIP alignment score, image-complexity correlations,
FID, duplication analysis and match-gallery plots; wandb-schema metrics
with an always-on JSONL fallback.

Distributed: launch under `python -m torch.distributed.run
--nproc-per-node N diff_retrieval.py ...` (one process per GPU over
RCCL; replaces the reference's mp.spawn path). Runs end-to-end on CPU
for the no-GPU plumbing config (BASELINE config 1).
"""
from __future__ import annotations

import argparse
import json
from pathlib import Path

import numpy as np
import torch
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

from dcr_amd.data import SynthDataset, EvalTransform
from dcr_amd.parallel import barrier, init_distributed_mode, is_main_process
from dcr_amd.retrieval import (einsum_in_chunks, extract_features,
                               gen_clipscore, glcm_entropy, jpeg_size,
                               l2_normalize, load_clip, load_dino, load_sscd,
                               pearson_with_p, sim_matrix,
                               similarity_histogram, top_matches, topk_stats,
                               tv_loss)
from dcr_amd.utils import Tracker


def parse_args():
    p = argparse.ArgumentParser("Generic image retrieval given a path")
    p.add_argument("--query_dir", type=str, required=True, help="The inferences")
    p.add_argument("--val_dir", type=str, required=True, help="The train data")
    p.add_argument("--pt_style", default="sscd", type=str)
    p.add_argument("-a", "--arch", default="resnet50", type=str)
    p.add_argument("-j", "--workers", default=4, type=int)
    p.add_argument("-b", "--batch-size", default=128, type=int)
    p.add_argument("--world-size", default=-1, type=int)
    p.add_argument("--rank", default=-1, type=int)
    p.add_argument("--dist-url", default="env://", type=str)
    p.add_argument("--dist-backend", default="nccl", type=str)
    p.add_argument("--seed", default=None, type=int)
    p.add_argument("--gpu", default=None, type=int)
    p.add_argument("--multiprocessing-distributed", action="store_true")
    p.add_argument("--multiscale", default=False, type=lambda s: s in ("1", "true", "True"))
    p.add_argument("--pretrained", default="", type=str)
    p.add_argument("--similarity_metric", default="dotproduct", type=str)
    p.add_argument("--num_loss_chunks", default=1, type=int,
                   help="(reference flag; splitloss here derives per-patch "
                        "descriptors from the model's own token/spatial map "
                        "instead of slicing the flat embedding)")
    p.add_argument("--numpatches", default=1, type=int,
                   help="(reference flag; see --num_loss_chunks)")
    p.add_argument("--isvit", action="store_true",
                   help="(reference flag; ViT-ness is detected from the model)")
    p.add_argument("--layer", default=1, type=int)
    p.add_argument("--stype", default="", type=str)
    p.add_argument("--keephead", action="store_true")
    p.add_argument("--keeppredictor", action="store_true")
    p.add_argument("-ssp", "--sim_save_path", type=str, default="./similarityscores/")
    p.add_argument("--einsum_chunks", default=30, type=int)
    p.add_argument("--dontsave", action="store_true")
    p.add_argument("--num_matches", default=4, type=int)
    p.add_argument("--imsize", default=224, type=int)
    p.add_argument("--noeval", action="store_true")
    p.add_argument("--skip_fid", action="store_true")
    p.add_argument("--ipr", action="store_true",
                   help="Improved Precision & Recall (reference imports IPR at "
                        "diff_retrieval.py:587 but left the call commented out)")
    p.add_argument("--project", default="imsimv2_retrieval", type=str)
    return p.parse_args()


# reference arch mappings (diff_retrieval.py:250-283)
_DINO_ARCH = {"vit_base": "dino_vitb16", "vit_base8": "dino_vitb8",
              "vit_small": "dino_vits16", "resnet50": "dino_resnet50"}
_CLIP_ARCH = {"vit_large": "ViT-L/14", "vit_base": "ViT-B/16",
              "resnet50": "RN50x16"}
_SSCD_ARCH = {"resnet50": "sscd", "resnet50_im": "sscd_im",
              "resnet50_disc": "sscd_disc_large"}


def build_backbone(args, device):
    """Backbone per (--pt_style, --arch), reference diff_retrieval.py:249-285.
    Our folded spellings (--pt_style dino_vits8 / sscd_im / ...) also work."""
    if args.pt_style.startswith("dino"):
        if args.pt_style != "dino":            # folded spelling
            arch = args.pt_style
        elif args.arch in _DINO_ARCH:
            arch = _DINO_ARCH[args.arch]
        elif args.arch.startswith("dino_"):
            arch = args.arch
        else:
            raise NotImplementedError(
                f"dino arch {args.arch!r}: supported are {sorted(_DINO_ARCH)} "
                "(reference's cifar10 variant is not built)")
        return load_dino(arch, weights=args.pretrained or None, device=device)
    if args.pt_style == "clip":
        model, _ = load_clip(_CLIP_ARCH.get(args.arch, "ViT-B/16"),
                             device=device)
        return lambda x: model.encode_image(x)
    # sscd: reference picks the torchscript file by --arch
    name = _SSCD_ARCH.get(args.arch, args.pt_style) \
        if args.pt_style == "sscd" else args.pt_style
    return load_sscd(name, device=device)


@torch.no_grad()
def _patch_features(model, loader, device, args):
    """[N, D, P] L2-normalized per-patch descriptors: ViT token features
    (get_intermediate_layers) or the CNN's final spatial map."""
    import torch.nn.functional as F_
    feats = []
    for batch in loader:
        imgs = batch[0].to(device)
        if hasattr(model, "get_intermediate_layers"):
            tok = model.get_intermediate_layers(imgs, n=args.layer)[0][:, 1:]
            f = tok.transpose(1, 2)                     # [N, D, P]
        elif hasattr(model, "backbone"):                # SSCDModel
            fm = model.backbone.forward_features(imgs)  # [N, D, h, w]
            f = fm.flatten(2)
        else:
            f = model(imgs)[..., None]
        feats.append(F_.normalize(f, dim=1).cpu())
    return torch.cat(feats)


def main():
    args = parse_args()
    if args.similarity_metric == "splitlosscross":
        # reference alias (diff_retrieval.py:188-190)
        args.similarity_metric = "splitloss"
        args.stype = "cross"
    rank, world, local = init_distributed_mode()
    device = torch.device("cuda", local) if torch.cuda.is_available() \
        else torch.device("cpu")
    if args.seed is not None:
        torch.manual_seed(args.seed)
        torch.backends.cudnn.deterministic = True

    model = build_backbone(args, device)

    tf = EvalTransform(args.imsize)
    query_ds = SynthDataset(args.query_dir, transform=tf, with_prompts=True)
    val_ds = SynthDataset(args.val_dir, transform=tf)
    print(f"query: {len(query_ds)} imgs, values: {len(val_ds)} imgs")

    def loader(ds):
        sampler = DistributedSampler(ds, shuffle=False) if world > 1 else None
        return DataLoader(ds, batch_size=min(64, max(1, args.batch_size // 2)),
                          sampler=sampler, num_workers=args.workers,
                          pin_memory=device.type == "cuda")

    # HOT LOOP 1: distributed feature extraction + all-gather
    query_f = extract_features(model, loader(query_ds), device,
                               multiscale=args.multiscale)
    val_f = extract_features(model, loader(val_ds), device,
                             multiscale=args.multiscale)
    barrier()
    if not is_main_process():
        return

    out_dir = Path(args.sim_save_path)
    out_dir.mkdir(parents=True, exist_ok=True)
    # run name = last 3 path components + metric (reference :378-382)
    dp = "/".join(Path(args.query_dir).parts[-3:])
    tracker = Tracker(args.project, name=f"{dp}_{args.similarity_metric}",
                      config=vars(args), out_dir=out_dir)

    query_f = l2_normalize(query_f)
    val_f = l2_normalize(val_f)

    # HOT LOOP 2: similarity
    if args.similarity_metric == "splitloss":
        # patch-wise max similarity (reference :393-400, einsum :643-662):
        # features here are [N, D, P] per-patch descriptors
        qp_ = _patch_features(model, loader(query_ds), device, args)
        vp_ = _patch_features(model, loader(val_ds), device, args)
        sim = einsum_in_chunks(qp_, vp_, chunk=max(1, len(query_ds) //
                                                   max(1, args.einsum_chunks)),
                               stype=args.stype)
        sim_tt = einsum_in_chunks(vp_, vp_, chunk=max(1, len(val_ds) //
                                                      max(1, args.einsum_chunks)),
                                  stype=args.stype)
    else:
        sim = sim_matrix(val_f, query_f).t()    # [n_gen, n_train] (rocBLAS)
        sim_tt = sim_matrix(val_f, val_f)       # [n_train, n_train]

    if not args.dontsave:
        torch.save(sim.cpu(), out_dir / "similarity.pth")
        torch.save(sim_tt.cpu(), out_dir / "similarity_wtrain.pth")

    stats = topk_stats(sim, sim_tt)
    tracker.log(stats)
    print(json.dumps(stats, indent=2))

    top1 = sim.max(dim=1).values.float().cpu()
    counts, bins = similarity_histogram(top1)
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        plt.figure()
        plt.hist(top1.numpy(), bins=50, range=(0, 1), alpha=0.7, label="gen->train")
        t2 = sim_tt.topk(2, dim=1).values[:, 1].float().cpu()
        plt.hist(t2.numpy(), bins=50, range=(0, 1), alpha=0.5, label="train->train")
        plt.legend()
        plt.xlabel("top-1 SSCD similarity")
        plt.savefig(out_dir / "similarity_hist.png", dpi=120)
        plt.close()
    except Exception as e:
        print(f"histogram plot skipped: {e}")

    # CLIP alignment score (reference :486-495) when prompts exist
    if query_ds.prompts and not args.noeval:
        clip_model, clip_tok = load_clip(device=device)
        imgs = torch.stack([query_ds[i][0] for i in range(len(query_ds))])
        cs = gen_clipscore(clip_model, clip_tok, imgs, query_ds.prompts,
                           device=device)
        rec = {"clipscore": cs.mean().item()}
        if val_ds.prompts:
            vimgs = torch.stack([val_ds[i][0] for i in range(len(val_ds))])
            csb = gen_clipscore(clip_model, clip_tok, vimgs, val_ds.prompts,
                                device=device)
            rec["clipscore_bg"] = csb.mean().item()
        tracker.log(rec)

    # complexity correlations (reference :498-540): "are SIMPLE TRAINING
    # images copied more?" — metrics are computed on each generation's
    # top-1 MATCHED TRAIN image, correlated with the match similarity.
    if not args.noeval:
        from PIL import Image
        match_idx = sim.argmax(dim=1).tolist()
        dbsims = top1.numpy()
        ent, crs, tv = [], [], []
        for loc in match_idx:
            img = Image.open(val_ds.files[loc]).convert("RGB")
            arr = np.asarray(img)
            ent.append(glcm_entropy(arr))
            crs.append(jpeg_size(img, quality=90) / 1024)   # KB, reference :524
            t = torch.from_numpy(arr.copy()).permute(2, 0, 1).float()
            tv.append(tv_loss(t).item())
        ent = np.array(ent); crs = np.array(crs); tv = np.array(tv)
        cc_ent, pval_ent = pearson_with_p(ent, dbsims)
        cc_comp, pval_comp = pearson_with_p(crs, dbsims)
        cc_tvl, pval_tvl = pearson_with_p(tv, dbsims)
        cc_mixed, pval_mixed = pearson_with_p(ent * crs ** 0.5, dbsims)
        tracker.log({
            "cc_ent": cc_ent, "pval_ent": pval_ent,
            "cc_comp": cc_comp, "pval_comp": pval_comp,
            "cc_tvl": cc_tvl, "pval_tvl": pval_tvl,
            "cc_mixed": cc_mixed, "pval_mixed": pval_mixed,
        })
        if not args.dontsave:
            # reference artifact names (:553-556)
            torch.save(ent, out_dir / "entropies.pth")
            torch.save(tv, out_dir / "totvar.pth")
            torch.save(crs, out_dir / "compressions.pth")
            torch.save(dbsims, out_dir / "dbsims.pth")
            try:
                import matplotlib
                matplotlib.use("Agg")
                import matplotlib.pyplot as plt
                # reference filenames (:544-559; the reference saves the
                # 4th plot over the 3rd — SURVEY §2.6.7 — we keep all)
                for name, vals, cc, pv in (
                        ("entropies", ent, cc_ent, pval_ent),
                        ("tvls", tv, cc_tvl, pval_tvl),
                        ("crs", crs, cc_comp, pval_comp),
                        ("mixed", ent * crs ** 0.5, cc_mixed, pval_mixed)):
                    plt.figure(figsize=(4, 4))
                    plt.scatter(vals, dbsims, s=8, alpha=0.6)
                    plt.xlabel("simplicity")
                    plt.ylabel("sims")
                    plt.title(f"CC={cc:.4f}, pval={pv:.3g}")
                    plt.tight_layout()
                    plt.savefig(out_dir / f"simplicityscatter_{name}.png", dpi=110)
                    plt.close()
            except Exception as e:
                print(f"scatter plots skipped: {e}")

    # duplication analysis (reference :562-583): are the up-weighted
    # (duplicated) training images matched more often / more strongly?
    if not args.noeval:
        import glob as _glob
        import pickle as _pickle
        wpicks = sorted(_glob.glob(str(Path(args.val_dir) / "weights_*.pickle")))
        if wpicks:
            with open(wpicks[0], "rb") as fh:
                weights = _pickle.load(fh)
            if len(weights) == sim.shape[1]:
                w = torch.tensor([float(x) for x in weights])
                dup_mask = w > 1
                match_idx = sim.argmax(dim=1)
                matched_dup = dup_mask[match_idx].float().mean().item()
                dup_stats = {
                    "dup_frac_of_train": dup_mask.float().mean().item(),
                    "dup_matched_frac": matched_dup,
                    "sim_to_dup_mean": sim[:, dup_mask].max(dim=1).values.mean().item()
                        if dup_mask.any() else 0.0,
                    "sim_to_nodup_mean": sim[:, ~dup_mask].max(dim=1).values.mean().item()
                        if (~dup_mask).any() else 0.0,
                }
                tracker.log(dup_stats)
                print(json.dumps(dup_stats, indent=2))

    # FID (reference :597-600; HOT LOOP 3)
    if not args.noeval and not args.skip_fid:
        from dcr_amd.metrics import calculate_fid_given_paths
        fid = calculate_fid_given_paths([args.query_dir, args.val_dir],
                                        batch_size=50, device=str(device),
                                        dims=2048)
        tracker.log({"fid": fid})
        print(f"FID: {fid:.3f}")

    # Improved Precision & Recall (VGG16 manifold; reference :587,601-605)
    if args.ipr and not args.noeval:
        from dcr_amd.metrics import IPR
        ipr = IPR(batch_size=16, k=3, device=str(device))
        ipr.compute_manifold_ref(args.val_dir)
        pr = ipr.precision_and_recall(args.query_dir)
        tracker.log({"precision": pr.precision, "recall": pr.recall})
        print(f"IPR precision={pr.precision:.4f} recall={pr.recall:.4f}")

    # gallery: top matches for the most-copied generations (reference :609-640)
    if not args.dontsave:
        try:
            import matplotlib
            matplotlib.use("Agg")
            import matplotlib.pyplot as plt
            from PIL import Image
            k = args.num_matches
            n_show = min(10, len(query_ds))
            order = torch.argsort(top1, descending=True)[:n_show]
            vals, idxs = top_matches(sim, k=k)
            fig, axes = plt.subplots(n_show, k + 1,
                                     figsize=(2 * (k + 1), 2 * n_show))
            if n_show == 1:
                axes = axes[None, :]
            for row, qi in enumerate(order.tolist()):
                axes[row, 0].imshow(Image.open(query_ds.files[qi]))
                axes[row, 0].set_title(f"gen {qi}", fontsize=6)
                for col in range(k):
                    ti = int(idxs[qi, col])
                    axes[row, col + 1].imshow(Image.open(val_ds.files[ti]))
                    axes[row, col + 1].set_title(f"{vals[qi, col]:.2f}", fontsize=6)
                for ax in axes[row]:
                    ax.axis("off")
            plt.tight_layout()
            plt.savefig(out_dir / "gallery.png", dpi=100)
            plt.close()
        except Exception as e:
            print(f"gallery plot skipped: {e}")

    tracker.finish()


if __name__ == "__main__":
    main()
