import os
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parents[1]
sys.path.insert(0, str(REPO))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X (HIP) GPU")


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no HIP GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tmp_sandbox(tmp_path):
    """Per-test scratch dir (the reference's tf-sandbox analog,
    SharedSparkSessionSuite.scala:29-43)."""
    return tmp_path


@pytest.fixture(scope="session")
def tf_example_protos():
    """Dynamically-built tensorflow.Example / SequenceExample message classes
    from google.protobuf — an independent implementation of the wire format
    used as the interop golden reference."""
    from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

    pool = descriptor_pool.DescriptorPool()
    fd = descriptor_pb2.FileDescriptorProto()
    fd.name = "tf_example_test.proto"
    fd.package = "tensorflow"
    fd.syntax = "proto3"

    def msg(name):
        m = fd.message_type.add()
        m.name = name
        return m

    b = msg("BytesList")
    f = b.field.add(); f.name = "value"; f.number = 1; f.label = 3; f.type = 12
    fl = msg("FloatList")
    f = fl.field.add(); f.name = "value"; f.number = 1; f.label = 3; f.type = 2
    f.options.packed = True
    il = msg("Int64List")
    f = il.field.add(); f.name = "value"; f.number = 1; f.label = 3; f.type = 3
    f.options.packed = True
    feat = msg("Feature")
    for nm, num, ty in [("bytes_list", 1, ".tensorflow.BytesList"),
                        ("float_list", 2, ".tensorflow.FloatList"),
                        ("int64_list", 3, ".tensorflow.Int64List")]:
        f = feat.field.add(); f.name = nm; f.number = num; f.label = 1; f.type = 11
        f.type_name = ty
    feats = msg("Features")
    f = feats.field.add(); f.name = "feature"; f.number = 1; f.label = 3; f.type = 11
    entry = feats.nested_type.add(); entry.name = "FeatureEntry"
    entry.options.map_entry = True
    k = entry.field.add(); k.name = "key"; k.number = 1; k.label = 1; k.type = 9
    v = entry.field.add(); v.name = "value"; v.number = 2; v.label = 1; v.type = 11
    v.type_name = ".tensorflow.Feature"
    f.type_name = ".tensorflow.Features.FeatureEntry"
    ex = msg("Example")
    f = ex.field.add(); f.name = "features"; f.number = 1; f.label = 1; f.type = 11
    f.type_name = ".tensorflow.Features"
    fls = msg("FeatureList")
    f = fls.field.add(); f.name = "feature"; f.number = 1; f.label = 3; f.type = 11
    f.type_name = ".tensorflow.Feature"
    flm = msg("FeatureLists")
    f = flm.field.add(); f.name = "feature_list"; f.number = 1; f.label = 3
    f.type = 11
    entry2 = flm.nested_type.add(); entry2.name = "FeatureListEntry"
    entry2.options.map_entry = True
    k = entry2.field.add(); k.name = "key"; k.number = 1; k.label = 1; k.type = 9
    v = entry2.field.add(); v.name = "value"; v.number = 2; v.label = 1; v.type = 11
    v.type_name = ".tensorflow.FeatureList"
    f.type_name = ".tensorflow.FeatureLists.FeatureListEntry"
    se = msg("SequenceExample")
    f = se.field.add(); f.name = "context"; f.number = 1; f.label = 1; f.type = 11
    f.type_name = ".tensorflow.Features"
    f = se.field.add(); f.name = "feature_lists"; f.number = 2; f.label = 1
    f.type = 11
    f.type_name = ".tensorflow.FeatureLists"
    pool.Add(fd)

    def cls(name):
        return message_factory.GetMessageClass(pool.FindMessageTypeByName(name))

    class Protos:
        Example = cls("tensorflow.Example")
        SequenceExample = cls("tensorflow.SequenceExample")
        Feature = cls("tensorflow.Feature")
        Features = cls("tensorflow.Features")
        FeatureList = cls("tensorflow.FeatureList")

    return Protos
