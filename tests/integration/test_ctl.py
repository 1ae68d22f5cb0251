"""amctl operator-CLI tests against the HTTP frontend."""
import yaml

from active_monitor_amd.cmd.ctl import build_parser
from active_monitor_amd.cmd.ctl import run as ctl_run

from .conftest import make_hc
from .test_http import HttpEnv


def test_ctl_lifecycle(run, capsys, tmp_path):
    async def go():
        async with HttpEnv() as env:
            await env.client.create(make_hc(name="ctl-check"))
            obj = await env.client.get(
                "activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health",
                "ctl-check")
            obj["status"] = {"status": "Succeeded", "successCount": 4,
                             "failedCount": 1}
            await env.client.update_status(obj)
            url = env.frontend.url

            async def ctl(*argv):
                rc = await ctl_run(build_parser().parse_args(["--server", url, *argv]))
                return rc, capsys.readouterr().out

            rc, out = await ctl("get", "hc", "-n", "health")
            assert rc == 0
            assert "NAME" in out and "LATEST STATUS" in out
            assert "ctl-check" in out and "Succeeded" in out
            line = next(l for l in out.splitlines() if l.startswith("ctl-check"))
            cols = line.split()
            assert cols[1] == "Succeeded" and cols[2] == "4" and cols[3] == "1"

            rc, out = await ctl("get", "hc", "ctl-check", "-n", "health", "-o", "yaml")
            assert rc == 0 and "repeatAfterSec" in out

            rc, out = await ctl("describe", "hc", "ctl-check", "-n", "health")
            assert rc == 0 and "successCount: 4" in out

            f = tmp_path / "hc.yaml"
            doc = make_hc(name="applied")
            f.write_text(yaml.safe_dump(doc))
            rc, out = await ctl("apply", "-f", str(f))
            assert rc == 0 and "healthcheck/applied created" in out
            doc["spec"]["repeatAfterSec"] = 77
            f.write_text(yaml.safe_dump(doc))
            rc, out = await ctl("apply", "-f", str(f))
            assert rc == 0 and "healthcheck/applied configured" in out
            fresh = await env.client.get(
                "activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health",
                "applied")
            assert fresh["spec"]["repeatAfterSec"] == 77

            rc, out = await ctl("delete", "hc", "applied", "-n", "health")
            assert rc == 0 and "deleted" in out

            rc, _ = await ctl("get", "hc", "applied", "-n", "health")
            assert rc == 1  # NotFound

            # workflow + events listings render
            await env.client.create({
                "apiVersion": "argoproj.io/v1alpha1", "kind": "Workflow",
                "metadata": {"name": "w1", "namespace": "health"},
                "spec": {}, "status": {"phase": "Running"},
            })
            rc, out = await ctl("get", "wf", "-n", "health")
            assert rc == 0 and "w1" in out and "Running" in out

    run(go(), timeout=60)
