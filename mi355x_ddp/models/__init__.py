from .resnet import ResNet, BasicBlock, Bottleneck, resnet18, resnet34, resnet50, build_model  # noqa: F401
