"""Regenerate docs/api.md from the live param system.

Usage: python tools/gen_api_doc.py
"""
import sys
import warnings

warnings.filterwarnings("ignore")
sys.path.insert(0, ".")

import spark_rapids_ml_amd as pkg  # noqa: E402

CLASSES = [
    "KMeans", "KMeansModel", "DBSCAN", "DBSCANModel", "PCA", "PCAModel",
    "LinearRegression", "LinearRegressionModel",
    "LogisticRegression", "LogisticRegressionModel",
    "RandomForestClassifier", "RandomForestClassificationModel",
    "RandomForestRegressor", "RandomForestRegressionModel",
    "NearestNeighbors", "ApproximateNearestNeighbors",
    "UMAP", "UMAPModel",
]

TAIL = """## Evaluators

`RegressionEvaluator` (rmse|mse|r2|mae|var), `MulticlassClassificationEvaluator` \
(f1|accuracy|weighted*|hammingLoss|logLoss + *ByLabel metrics via `metricLabel`), \
`BinaryClassificationEvaluator` (areaUnderROC|areaUnderPR).

## Tuning & Pipeline

`CrossValidator(estimator, estimatorParamMaps, evaluator, numFolds, seed, \
collectSubModels)` with single-pass `fitMultiple` and save/load; \
`ParamGridBuilder`; `Pipeline`/`PipelineModel` with the VectorAssembler \
bypass and save/load; `VectorAssembler`, `NoOpTransformer`.

## CLI entry points

| command | purpose |
|---|---|
| `srml-amd-run app.py` (= `python -m spark_rapids_ml_amd app.py`) | run a reference-style script unmodified: installs the `spark_rapids_ml` aliases, initializes the communicator |
| `srml-amd-launch [--gpus N] app.py` | spark-submit analog: execs torchrun with one rank per visible GPU, routing the script through the runner above |
| `srml-amd-server --port 8571` | remote fit/transform HTTP service (Spark Connect plugin analog) |
| `srml-amd-install` | print/emit the no-import-change alias setup |

All estimators, models, `Pipeline`/`PipelineModel`, `CrossValidator`/
`CrossValidatorModel` support `save(path)` / `.load(path)` (and
`.write().overwrite().save()`); models persist as `metadata.json` +
`attributes.npz` (+`attributes.json`).
"""


def fmt_default(inst, p):
    if not inst.hasDefault(p):
        return "—"
    v = inst.getOrDefault(p.name)
    return "None" if v is None else str(v)


def main() -> None:
    lines = ["# API reference (generated from the live param system)", ""]
    for name in CLASSES:
        cls = getattr(pkg, name)
        lines.append(f"## {name}")
        lines.append("")
        doc = (cls.__doc__ or "").strip().splitlines()
        if doc:
            lines.append(doc[0].strip())
            lines.append("")
        try:
            inst = cls()
        except Exception:
            inst = None
        if inst is not None and inst.params:
            mapping = cls._param_mapping() if hasattr(cls, "_param_mapping") else {}
            lines.append("| Spark param | default | native param |")
            lines.append("|---|---|---|")
            for p in inst.params:
                native = mapping.get(p.name, p.name)
                if native is None:
                    native = "*(unsupported: raises)*"
                elif native == "":
                    native = "*(accepted, no native effect)*"
                lines.append(f"| {p.name} | {fmt_default(inst, p)} | {native} |")
            lines.append("")
    lines.append(TAIL)
    with open("docs/api.md", "w") as f:
        f.write("\n".join(lines))
    print("docs/api.md regenerated")


if __name__ == "__main__":
    main()
