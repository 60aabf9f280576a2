"""RUNME — deploy/run the example workloads as a dependency-ordered job
(the `group_apply/RUNME.py` Workflow equivalent: 4 tasks, 01→02 deps,
28800 s job timeout)."""
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.utils.jobs import Job, Task

EX = __file__.rsplit("/", 1)[0]


def main():
    job = Job("mi355x-scale-examples", [
        Task("01_fine_grained_forecasting",
             [sys.executable, f"{EX}/01_fine_grained_forecasting.py"]),
        Task("02_hyperopt",
             [sys.executable, f"{EX}/02_hyperopt.py"],
             depends_on=["01_fine_grained_forecasting"]),
        Task("03_hyperopt_data_sizes",
             [sys.executable, f"{EX}/03_hyperopt_data_sizes.py"],
             depends_on=["02_hyperopt"]),
        Task("04_distributed_training",
             [sys.executable, f"{EX}/04_distributed_training.py"],
             depends_on=["01_fine_grained_forecasting"]),
    ], timeout_seconds=28800, max_concurrent_tasks=2)
    results = job.run()
    for k, r in results.items():
        print(f"{k:32s} {r.status:8s} {r.seconds:7.1f}s")
    if any(r.status != "SUCCESS" for r in results.values()):
        sys.exit(1)


if __name__ == "__main__":
    main()
