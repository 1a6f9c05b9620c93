"""CLI parsing tests.

The flag surface mirrors the reference's clap CLI (reference
gpu-pruner/src/main.rs:46-134): same names, shorts, value enums, defaults —
plus the MI355X-native --max-concurrency / --queue-capacity knobs.
"""


def parse(core, *args):
    return core.parse_cli(list(args))


def test_defaults(core):
    r = parse(core, "--prometheus-url", "http://prom:9090")
    assert r["error"] is None
    c = r["config"]
    assert c["duration"] == 30
    assert c["daemon_mode"] is False
    assert c["enabled_resources"] == "drsin"
    assert c["check_interval"] == 180
    assert c["namespace"] is None
    assert c["grace_period"] == 300
    assert c["model_name"] is None
    assert c["power_threshold"] is None
    assert c["honor_labels"] is False
    assert c["run_mode"] == "dry-run"
    assert c["prometheus_tls_mode"] == "verify"
    assert c["log_format"] == "default"


def test_all_long_flags(core):
    r = parse(core,
              "--duration", "45", "--daemon-mode", "--enabled-resources", "dn",
              "--check-interval", "60", "--namespace", "ml-.*",
              "--grace-period", "120", "--model-name", "AMD Instinct MI355X",
              "--power-threshold", "150", "--honor-labels", "true",
              "--run-mode", "scale-down", "--prometheus-url", "https://prom:9091",
              "--prometheus-token", "tok", "--prometheus-tls-mode", "skip",
              "--prometheus-tls-cert", "/certs/ca.pem", "--log-format", "json",
              "--max-concurrency", "64", "--queue-capacity", "200")
    assert r["error"] is None
    c = r["config"]
    assert c["duration"] == 45
    assert c["daemon_mode"] is True
    assert c["enabled_resources"] == "dn"
    assert c["check_interval"] == 60
    assert c["namespace"] == "ml-.*"
    assert c["grace_period"] == 120
    assert c["model_name"] == "AMD Instinct MI355X"
    assert c["power_threshold"] == 150.0
    assert c["honor_labels"] is True
    assert c["run_mode"] == "scale-down"
    assert c["prometheus_tls_mode"] == "skip"
    assert c["log_format"] == "json"
    assert c["max_concurrency"] == 64
    assert c["queue_capacity"] == 200


def test_short_flags(core):
    r = parse(core, "-t", "15", "-d", "-e", "i", "-c", "30", "-n", "team-a",
              "-g", "60", "-m", "MI355X", "-r", "scale-down",
              "-l", "pretty", "--prometheus-url", "http://p")
    assert r["error"] is None
    c = r["config"]
    assert c["duration"] == 15
    assert c["daemon_mode"] is True
    assert c["enabled_resources"] == "i"
    assert c["check_interval"] == 30
    assert c["namespace"] == "team-a"
    assert c["grace_period"] == 60
    assert c["model_name"] == "MI355X"
    assert c["run_mode"] == "scale-down"
    assert c["log_format"] == "pretty"


def test_equals_style(core):
    r = parse(core, "--prometheus-url=http://p", "--duration=7", "--honor-labels=false")
    assert r["error"] is None
    assert r["config"]["duration"] == 7
    assert r["config"]["honor_labels"] is False


def test_honor_labels_bare_sets_true(core):
    r = parse(core, "--honor-labels", "--prometheus-url", "http://p")
    assert r["error"] is None
    assert r["config"]["honor_labels"] is True


def test_prometheus_url_required(core):
    r = parse(core)
    assert r["error"] is not None and "prometheus-url" in r["error"]


def test_unknown_flag_rejected(core):
    r = parse(core, "--prometheus-url", "http://p", "--bogus")
    assert r["error"] is not None and "--bogus" in r["error"]


def test_invalid_run_mode_rejected(core):
    r = parse(core, "--prometheus-url", "http://p", "--run-mode", "chaos")
    assert r["error"] is not None


def test_invalid_log_format_rejected(core):
    r = parse(core, "--prometheus-url", "http://p", "--log-format", "xml")
    assert r["error"] is not None


def test_help(core):
    r = parse(core, "--help")
    assert r["help"] is True


def test_leader_elect_flags(core):
    r = parse(core, "--prometheus-url", "http://p", "--leader-elect",
              "--leader-elect-lease-duration", "30",
              "--leader-elect-renew-period", "10")
    assert r["config"]["leader_elect"] is True
    assert r["config"]["leader_lease_duration_s"] == 30
    assert r["config"]["leader_renew_period_s"] == 10


def test_leader_elect_defaults(core):
    r = parse(core, "--prometheus-url", "http://p")
    assert r["config"]["leader_elect"] is False
    assert r["config"]["leader_lease_duration_s"] == 15
    assert r["config"]["leader_renew_period_s"] == 5


def test_help_text_covers_new_flags():
    import subprocess
    from pathlib import Path

    binary = Path(__file__).resolve().parent.parent / "bin" / "gpu-pruner"
    r = subprocess.run([str(binary), "--help"], capture_output=True, text=True,
                       timeout=30)
    assert r.returncode == 0
    for flag in ("--leader-elect", "--eval-strategy", "--metrics-port",
                 "--max-concurrency", "--honor-labels", "--power-threshold"):
        assert flag in r.stdout, flag
