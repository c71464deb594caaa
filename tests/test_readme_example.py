"""The README quickstart must actually run (doc-rot guard)."""

import numpy as np
import torch


def test_readme_quickstart(tmp_path):
    # mirror of the README snippet, on a synthetic CSV
    rows = ["protocol_type,service,flag,duration,src_bytes,label"]
    r = np.random.default_rng(0)
    protos = ["tcp", "udp", "icmp"]
    services = ["http", "dns", "ftp"]
    flags = ["SF", "REJ"]
    for i in range(600):
        p = protos[r.integers(0, 3)]
        s = services[r.integers(0, 3)]
        f = flags[r.integers(0, 2)]
        d = float(r.uniform(0, 10))
        b = float(r.uniform(0, 1000))
        lab = "attack" if (p == "udp") ^ (b > 500) else "normal"
        rows.append(f"{p},{s},{f},{d:.3f},{b:.1f},{lab}")
    csv = tmp_path / "flows.csv"
    csv.write_text("\n".join(rows) + "\n")

    from flowml import Pipeline, read_csv
    from flowml.feature import StringIndexer, OneHotEncoder, VectorAssembler
    from flowml.classification import RandomForestClassifier
    from flowml.evaluation import MulticlassClassificationEvaluator

    df = read_csv(str(csv))
    train, test = df.randomSplit([0.8, 0.2], seed=42)
    pipe = Pipeline(stages=[
        StringIndexer(inputCols=["protocol_type", "service", "flag"],
                      outputCols=["p_idx", "s_idx", "f_idx"]),
        OneHotEncoder(inputCols=["p_idx", "s_idx", "f_idx"],
                      outputCols=["p_vec", "s_vec", "f_vec"]),
        VectorAssembler(inputCols=["p_vec", "s_vec", "f_vec", "duration", "src_bytes"],
                        outputCol="features"),
        StringIndexer(inputCol="label", outputCol="labelIdx"),
        RandomForestClassifier(numTrees=20, labelCol="labelIdx"),
    ])
    model = pipe.fit(train)
    f1 = MulticlassClassificationEvaluator(labelCol="labelIdx").evaluate(model.transform(test))
    assert f1 > 0.8
    model.save(str(tmp_path / "model_dir"))
    from flowml import PipelineModel

    loaded = PipelineModel.load(str(tmp_path / "model_dir"))
    a = loaded.transform(test)["prediction"]
    b = model.transform(test)["prediction"]
    assert torch.equal(a, b)
