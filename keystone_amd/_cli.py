"""CLI implementation (see bin/run_pipeline.py for usage)."""


import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser(description=__doc__)
    sub = p.add_subparsers(dest="app", required=True)

    ng = sub.add_parser("newsgroups")
    ng.add_argument("--n-docs", type=int, default=2000)
    ng.add_argument("--num-classes", type=int, default=20)
    ng.add_argument("--common-features", type=int, default=20000)
    ng.add_argument("--data-dir", help="dir-per-class text corpus")

    am = sub.add_parser("amazon")
    am.add_argument("--n-docs", type=int, default=2000)
    am.add_argument("--data-json", help="JSON-lines reviews file")

    mn = sub.add_parser("mnist")
    mn.add_argument("--n", type=int, default=10000)
    mn.add_argument("--features", choices=["cosine", "fft"], default="cosine")
    mn.add_argument("--num-features", type=int, default=4096)

    tm = sub.add_parser("timit")
    tm.add_argument("--n", type=int, default=50000)
    tm.add_argument("--num-features", type=int, default=8192)
    tm.add_argument("--epochs", type=int, default=3)

    cf = sub.add_parser("cifar")
    cf.add_argument("--n", type=int, default=2000)
    cf.add_argument("--num-filters", type=int, default=256)
    cf.add_argument("--data-bin", help="CIFAR-10 binary file")

    vc = sub.add_parser("voc")
    vc.add_argument("--images", type=int, default=128)
    vc.add_argument("--image-size", type=int, default=128)
    vc.add_argument("--num-classes", type=int, default=20)
    vc.add_argument("--data-dir", help="VOC-style dir with index.json")

    im = sub.add_parser("imagenet-fv")
    im.add_argument("--images", type=int, default=128)
    im.add_argument("--image-size", type=int, default=256)
    im.add_argument("--num-classes", type=int, default=1000)
    im.add_argument("--gmm-k", type=int, default=64)
    im.add_argument("--save", help="save the fitted pipeline here")
    im.add_argument("--solver-checkpoint", help="epoch checkpoint dir "
                    "(re-run with the same dir to resume the solver)")

    args = p.parse_args()

    from keystone_amd.parallel.runtime import init_runtime
    rt = init_runtime()
    t0 = time.time()

    if args.app == "newsgroups":
        from keystone_amd.evaluation import MulticlassClassifierEvaluator
        from keystone_amd.loaders import NewsgroupsDataLoader, synthetic_documents
        from keystone_amd.models import NewsgroupsConfig, build_newsgroups
        if args.data_dir:
            docs, labels, classes = NewsgroupsDataLoader(args.data_dir)
            k = len(classes)
        else:
            docs, labels = synthetic_documents(args.n_docs, args.num_classes)
            k = args.num_classes
        pipe = build_newsgroups(docs, labels, NewsgroupsConfig(
            num_classes=k, common_features=args.common_features))
        m = MulticlassClassifierEvaluator(k).evaluate(pipe.apply(docs), labels)
        if rt.rank == 0:
            print(f"train {m.summary()}  ({time.time()-t0:.1f}s)")

    elif args.app == "amazon":
        from keystone_amd.evaluation import BinaryClassifierEvaluator
        from keystone_amd.loaders import AmazonReviewsDataLoader, synthetic_documents
        from keystone_amd.models import AmazonConfig, build_amazon
        if args.data_json:
            docs, labels = AmazonReviewsDataLoader(args.data_json)
        else:
            docs, labels = synthetic_documents(args.n_docs, 2)
        pipe = build_amazon(docs, labels, AmazonConfig())
        m = BinaryClassifierEvaluator().evaluate(pipe.apply(docs), labels)
        if rt.rank == 0:
            print(f"train accuracy {m.accuracy:.4f}  ({time.time()-t0:.1f}s)")

    elif args.app == "mnist":
        from keystone_amd.evaluation import MulticlassClassifierEvaluator
        from keystone_amd.loaders import synthetic_mnist
        from keystone_amd.models import (MnistConfig, build_mnist_random_cosine,
                                         build_mnist_randomfft)
        train, labels = synthetic_mnist(args.n)
        conf = MnistConfig()
        pipe = (build_mnist_random_cosine(train, labels, conf,
                                          num_features=args.num_features)
                if args.features == "cosine"
                else build_mnist_randomfft(train, labels, conf))
        m = MulticlassClassifierEvaluator(10).evaluate(pipe.apply(train), labels)
        if rt.rank == 0:
            print(f"train {m.summary()}  ({time.time()-t0:.1f}s)")

    elif args.app == "timit":
        from keystone_amd.evaluation import MulticlassClassifierEvaluator
        from keystone_amd.loaders import synthetic_timit
        from keystone_amd.models import TimitConfig, build_timit
        train, labels = synthetic_timit(args.n)
        pipe = build_timit(train, labels, TimitConfig(
            num_features=args.num_features, num_iter=args.epochs))
        m = MulticlassClassifierEvaluator(147).evaluate(pipe.apply(train), labels)
        if rt.rank == 0:
            print(f"train {m.summary()}  ({time.time()-t0:.1f}s)")

    elif args.app == "cifar":
        import torch
        from keystone_amd.evaluation import MulticlassClassifierEvaluator
        from keystone_amd.loaders import CifarLoader
        from keystone_amd.models import CifarConfig, fit_cifar_patches
        import keystone_amd as ka
        if args.data_bin:
            ld = CifarLoader(args.data_bin)
            imgs, labels = ld.data, ld.labels
        else:
            g = torch.Generator().manual_seed(1)
            X = torch.rand(args.n, 3, 32, 32, generator=g)
            y = torch.randint(0, 10, (args.n,), generator=g)
            # class-dependent mean shift makes labels learnable
            X = (X + 0.25 * y.view(-1, 1, 1, 1) / 10).clamp(0, 1)
            imgs, labels = ka.from_rows(X.to(rt.device)), \
                ka.from_rows(y.to(rt.device))
        pipe = fit_cifar_patches(imgs, labels, CifarConfig(
            num_filters=args.num_filters))
        m = MulticlassClassifierEvaluator(10).evaluate(pipe.apply(imgs),
                                                       labels)
        if rt.rank == 0:
            print(f"train {m.summary()}  ({time.time()-t0:.1f}s)")

    elif args.app == "voc":
        import torch
        from keystone_amd.evaluation import MeanAveragePrecisionEvaluator
        from keystone_amd.loaders import VOCLoader, synthetic_images
        from keystone_amd.models import VOCConfig, fit_voc
        import keystone_amd as ka
        if args.data_dir:
            imgs, ml = VOCLoader(args.data_dir, args.num_classes)
        else:
            imgs, y = synthetic_images(args.images, h=args.image_size,
                                       w=args.image_size,
                                       num_classes=args.num_classes,
                                       class_structure=True)
            # single-positive multi-label matrix from the int labels
            import torch as _t
            M = _t.zeros(y.data.shape[0], args.num_classes,
                         device=y.data.device)
            M[_t.arange(y.data.shape[0]), y.data] = 1.0
            ml = ka.from_rows(M)
        pipe = fit_voc(imgs, ml, VOCConfig(num_classes=args.num_classes))
        scores = pipe.apply(imgs)
        ev = MeanAveragePrecisionEvaluator(args.num_classes).evaluate(
            scores, ml)
        if rt.rank == 0:
            print(f"train mAP {ev['mAP']:.4f}  ({time.time()-t0:.1f}s)")

    elif args.app == "imagenet-fv":
        from keystone_amd.evaluation import MulticlassClassifierEvaluator
        from keystone_amd.loaders import synthetic_images
        from keystone_amd.models import ImageNetFVConfig, fit_imagenet_fv
        imgs, labels = synthetic_images(args.images, h=args.image_size,
                                        w=args.image_size,
                                        num_classes=args.num_classes)
        conf = ImageNetFVConfig(num_classes=args.num_classes,
                                gmm_k=args.gmm_k,
                                solver_checkpoint_dir=args.solver_checkpoint)
        pipe, info = fit_imagenet_fv(imgs, labels, conf)
        if args.save:
            pipe.save(args.save)
        import keystone_amd as ka
        # the pipeline ends in TopKClassifier: column 0 is the top-1 class
        pred = pipe.apply(imgs).get().data[:, 0]
        m = MulticlassClassifierEvaluator(args.num_classes).evaluate(
            ka.from_rows(pred), labels)
        if rt.rank == 0:
            print(f"fit {info['t_total']:.2f}s, feature_dim "
                  f"{info['feature_dim']}, train accuracy "
                  f"{m.total_accuracy:.3f}  ({time.time()-t0:.1f}s total)")


if __name__ == "__main__":
    main()
