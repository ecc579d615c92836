#!/usr/bin/env bash
# Elasticity walkthrough (parity with the reference's demo scenario,
# scripts/run_demo_scenario.sh + walkthrough.md there): start the
# parser->detector pipeline, feed traffic, kill the detector mid-run
# (parser keeps running, drops are counted), restart it (late binding
# delivers buffered frames), then verify an anomaly alert end-to-end.
set -euo pipefail
cd "$(dirname "$0")/.."
WORK=$(mktemp -d)
trap 'kill $(jobs -p) 2>/dev/null || true; rm -rf "$WORK"' EXIT

python - "$WORK" <<'PY'
import subprocess, sys, time, yaml, os, signal
from pathlib import Path
sys.path.insert(0, os.getcwd())
from detectmateservice_amd.engine.sockets import PairDialer, PairListener, RecvTimeout
from detectmateservice_amd.schemas import LogSchema, DetectorSchema
from detectmateservice_amd.utils.synthetic import AUDIT_LOG_FORMAT, AUDIT_TEMPLATES, AuditLogGenerator
import httpx

work = Path(sys.argv[1])
parser_in = f"ipc://{work}/p.ipc"; det_in = f"ipc://{work}/d.ipc"; sink = f"ipc://{work}/s.ipc"
(work/"tpl.txt").write_text("\n".join(AUDIT_TEMPLATES))
def w(name, d): p = work/name; p.write_text(yaml.safe_dump(d)); return str(p)
ps = w("ps.yaml", {"component_type":"MatcherParser","engine_addr":parser_in,"out_addr":[det_in],
  "http_port":18111,"config_file":w("pc.yaml",{"parsers":{"MatcherParser":{
  "log_format":AUDIT_LOG_FORMAT,"params":{"path_templates":str(work/"tpl.txt")}}}}),
  "log_dir":str(work/"logs")})
ds = w("ds.yaml", {"component_type":"NewValueDetector","engine_addr":det_in,"out_addr":[sink],
  "http_port":18112,"config_file":w("dc.yaml",{"detectors":{"NewValueDetector":{
  "data_use_training":5,"global":{"g":{"header_variables":[{"pos":"Type"}]}}}}}),
  "log_dir":str(work/"logs")})

def start(settings):
    return subprocess.Popen([sys.executable,"-m","detectmateservice_amd.cli","--settings",settings],
                            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
def wait_up(port):
    for _ in range(100):
        try:
            if httpx.get(f"http://127.0.0.1:{port}/admin/status", timeout=1).json()["status"]["running"]:
                return True
        except Exception: time.sleep(0.2)
    return False

print("=> starting parser + detector services")
p1, p2 = start(ps), start(ds)
assert wait_up(18111) and wait_up(18112)
sk = PairListener(sink); feeder = PairDialer(parser_in); assert feeder.wait_connected(10)
gen = AuditLogGenerator(seed=1)
for i in range(5):
    feeder.send(LogSchema(logID=f"t{i}", log=gen.line()[0]).serialize())
time.sleep(1)

print("=> killing the detector mid-run (parser must keep running)")
p2.terminate(); p2.wait()
for i in range(20):
    feeder.send(LogSchema(logID=f"down{i}", log=gen.line()[0]).serialize())
time.sleep(1)
st = httpx.get("http://127.0.0.1:18111/admin/status", timeout=2).json()
assert st["status"]["engine_running"], "parser engine died with the detector down!"
print("   parser still running:", st["status"]["engine_running"])

print("=> restarting the detector (late binding reconnects)")
p2 = start(ds)
assert wait_up(18112)
time.sleep(1)
bad = "type=EVIL_DEMO msg=audit(1.0:1): pid=1 uid=0 auid=1 ses=1 msg='op=PAM:x acct=\"x\" exe=/bin/x hostname=? addr=? terminal=x res=success'"
feeder.send(LogSchema(logID="bad", log=bad).serialize())
# frames buffered while the detector was down are delivered on reconnect
# (late binding) and may alert first — drain until our sentinel arrives
seen = []
for _ in range(50):
    alert = DetectorSchema.deserialize(sk.recv(timeout_ms=15000))
    seen.append(alert.description)
    if "EVIL_DEMO" in alert.description:
        break
print(f"   {len(seen)} alert(s); buffered-frame alerts recovered: {len(seen)-1}")
print("   final alert:", seen[-1])
assert "EVIL_DEMO" in seen[-1]
print("=> demo complete: pipeline survived a stage death and recovered")
for port in (18111, 18112):
    try: httpx.post(f"http://127.0.0.1:{port}/admin/shutdown", timeout=2)
    except Exception: pass
for proc in (p1, p2):
    try:
        proc.wait(timeout=10)
    except subprocess.TimeoutExpired:
        proc.terminate()
        try:
            proc.wait(timeout=5)
        except subprocess.TimeoutExpired:
            proc.kill()
feeder.close(); sk.close()
PY
echo "OK"
