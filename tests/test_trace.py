import io

import pytest

from gpuschedule_amd import JobSpec, load_trace, save_trace, make_trace
from gpuschedule_amd.trace.synth import make_elastic_trace


def test_roundtrip():
    specs = make_trace(20, mix="philly-slice", seed=3)
    buf = io.StringIO()
    save_trace(specs, buf)
    buf.seek(0)
    loaded = load_trace(buf)
    assert loaded == specs


def test_loader_validates():
    with pytest.raises(ValueError):
        load_trace(io.StringIO("job_id,submit_time\n"))
    dup = ("job_id,submit_time,num_gpus,iterations,model\n"
           "a,0,1,10,resnet18\na,1,1,10,resnet18\n")
    with pytest.raises(ValueError):
        load_trace(io.StringIO(dup))


def test_loader_defaults_and_sort():
    csv = ("job_id,submit_time,num_gpus,iterations,model\n"
           "b,5.0,2,100,bert-base\n"
           "a,1.0,1,50,resnet18\n")
    specs = load_trace(io.StringIO(csv))
    assert [s.job_id for s in specs] == ["a", "b"]
    assert specs[0].elastic is False and specs[0].min_gpus == 1


def test_synth_deterministic():
    assert make_trace(10, seed=7) == make_trace(10, seed=7)
    assert make_trace(10, seed=7) != make_trace(10, seed=8)


def test_elastic_trace_shape():
    specs = make_elastic_trace()
    assert specs[0].elastic and specs[0].max_gpus == 8


def test_jobspec_validation():
    with pytest.raises(ValueError):
        JobSpec(job_id="x", submit_time=0, n_gpus=0, total_iters=1,
                model="resnet18")
