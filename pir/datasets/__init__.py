"""PIR-degradable evaluation workloads.

Each module exposes
    initialize(...) -> dataset object with
        .num_entries            table size the PIR plan covers
        .train_patterns         access traces for plan construction
        .eval_patterns          held-out traces
        .train_model(...)       train the task model (checkpointed)
        .evaluate(optimizer)    task metric under a BatchPIROptimize plan

The reference builds these on WikiText-2 / MovieLens-20M / Taobao CSVs
(paper/experimental/batch_pir/modules/*); this environment has no network
or datasets, so every loader takes an optional local data path and falls
back to a seeded synthetic generator of the same shape.
"""
