"""Generate a random-init model checkpoint in TF format (frozen .pb or
SavedModel) with our own proto writer — the no-TF equivalent of the
reference's freeze/export tooling.

  python tools/make_model.py resnet50 /tmp/r50.pb
  python tools/make_model.py lenet /tmp/lenet_sm --savedmodel
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("model", choices=["lenet", "resnet50", "inception_v3", "mobilenet_v2",
                                     "vit_b16"])
    p.add_argument("out")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--savedmodel", action="store_true",
                   help="emit a SavedModel dir (variables in a "
                        "TensorBundle) instead of a frozen .pb")
    args = p.parse_args()

    from tensorflow_web_deploy_amd import models as M
    builders = {
        "lenet": M.lenet.build_lenet,
        "resnet50": M.resnet50.build_resnet50,
        "inception_v3": __import__(
            "tensorflow_web_deploy_amd.models.inception_v3",
            fromlist=["x"]).build_inception_v3,
        "vit_b16": __import__(
            "tensorflow_web_deploy_amd.models.vit",
            fromlist=["x"]).build_vit_b16,
        "mobilenet_v2": __import__(
            "tensorflow_web_deploy_amd.models.mobilenet_v2",
            fromlist=["x"]).build_mobilenet_v2,
    }
    g = builders[args.model](seed=args.seed)
    if args.savedmodel:
        from tensorflow_web_deploy_amd.ckpt.savedmodel import save_savedmodel
        save_savedmodel(g, args.out, ["input"], ["softmax"])
        print(f"SavedModel written to {args.out}/")
    else:
        from tensorflow_web_deploy_amd.ckpt import graphdef as gd
        gd.save_frozen_graph(g, args.out)
        print(f"frozen graph written to {args.out} "
              f"({os.path.getsize(args.out) / 1e6:.1f} MB)")


if __name__ == "__main__":
    main()
