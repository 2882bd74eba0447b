from rocps.models.mnist import MnistMLP, mnist_mlp, synthetic_mnist_batch  # noqa: F401


def build_model(name: str, **kw):
    name = name.lower()
    if name in ("mnist", "mnist_mlp"):
        return mnist_mlp(**kw)
    if name in ("resnet50", "resnet-50"):
        from rocps.models.resnet import resnet50
        return resnet50(**kw)
    if name in ("wide_resnet101", "wide_resnet101_2", "wresnet101"):
        from rocps.models.resnet import wide_resnet101_2
        return wide_resnet101_2(**kw)
    if name in ("bert", "bert_base", "bert-base"):
        from rocps.models.bert import bert_base
        return bert_base(**kw)
    raise ValueError(f"unknown model {name!r}")
