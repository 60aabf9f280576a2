from mi355x_scale.utils.jobs import Job, Task


def test_job_dag_order_and_failure_skip():
    log = []

    def mk(name, fail=False):
        def _r():
            log.append(name)
            if fail:
                raise RuntimeError("boom")
            return name
        return _r

    job = Job("test", [
        Task("a", mk("a")),
        Task("b", mk("b", fail=True), depends_on=["a"]),
        Task("c", mk("c"), depends_on=["b"]),     # skipped (b failed)
        Task("d", mk("d"), depends_on=["a"]),     # still runs
    ], timeout_seconds=30)
    res = job.run()
    assert res["a"].status == "SUCCESS"
    assert res["b"].status == "FAILED"
    assert res["c"].status == "SKIPPED"
    assert res["d"].status == "SUCCESS"
    assert log.index("a") < log.index("b")
    assert "c" not in log


def test_job_subprocess_task():
    job = Job("sp", [Task("echo", ["python", "-c", "print('hi')"])],
              timeout_seconds=60)
    res = job.run()
    assert res["echo"].status == "SUCCESS"
    assert "hi" in res["echo"].output
