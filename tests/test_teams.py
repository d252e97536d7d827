"""Team distribution pipeline tests: local agents source -> render -> apply
with per-team prune, seeds, layered secrets, image registration."""

import pytest

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api
from kukeon_amd.controller.core import Controller
from kukeon_amd.runtime.process import FakeRuntime
from kukeon_amd.teams import parse_team_file
from kukeon_amd.teams.pipeline import build_order, team_init


@pytest.fixture
def agents_src(tmp_path):
    src = tmp_path / "agents"
    (src / "roles").mkdir(parents=True)
    (src / "harnesses").mkdir()
    (src / "roles" / "dev.yaml").write_text("""
apiVersion: kuketeams.io/v1
kind: Role
metadata: {name: dev}
spec:
  description: developer agent
  prompt: "You are the dev."
""")
    (src / "roles" / "pm.yaml").write_text("""
apiVersion: kuketeams.io/v1
kind: Role
metadata: {name: pm}
spec: {prompt: "You are the PM."}
""")
    (src / "harnesses" / "claude.yaml").write_text("""
apiVersion: kuketeams.io/v1
kind: Harness
metadata: {name: claude}
spec:
  skillPath: /skills
  makeTarget: claude-image
  template: harnesses/claude.tmpl.yaml
  seeds:
    - path: state/${HARNESS}/settings.json
      mode: 384
      content: '{"seeded": true}'
""")
    (src / "harnesses" / "claude.tmpl.yaml").write_text("""
kind: CellBlueprint
metadata: {name: ${TEAM}-${ROLE}-${HARNESS}}
spec:
  namePrefix: ${ROLE}
  template:
    spec:
      realmId: default
      spaceId: default
      stackId: default
      containers:
        - id: agent
          image: "${IMAGE}"
          command: /bin/sh
          args: ["-c", "echo ${ROLE_PROMPT}"]
""")
    (src / "catalog.yaml").write_text("""
apiVersion: kuketeams.io/v1
kind: ImageCatalog
spec:
  images:
    - ref: base
      harness: claude
      image: kukeon.internal/base
      capabilities: [root]
      build: {context: images/base, dockerfile: Dockerfile}
    - ref: go
      harness: claude
      base: base
      image: kukeon.internal/go
      capabilities: [root, go]
      build: {context: images/go, dockerfile: Dockerfile}
""")
    return src


def write_team_file(tmp_path, src, roles="[{ref: dev, needs: {image: [go]}}, {ref: pm}]"):
    tf = tmp_path / "kuketeam.yaml"
    tf.write_text(f"""
apiVersion: kuketeams.io/v1
kind: ProjectTeam
metadata: {{name: myteam}}
spec:
  source: {{path: {src}}}
  defaults: {{harnesses: [claude]}}
  roles: {roles}
""")
    return tf


def test_team_init_end_to_end(tmp_path, agents_src):
    ctl = Controller(str(tmp_path / "run"), runtime=FakeRuntime(),
                     gpu_devices=[])
    ctl.bootstrap()
    teams_root = tmp_path / "teams"
    (teams_root).mkdir()
    (teams_root / "secrets.env").write_text("GLOBAL_KEY=g1\nSHARED=host\n")
    (teams_root / "myteam").mkdir()
    (teams_root / "myteam" / "secrets.env").write_text("SHARED=team\n")
    tf = write_team_file(tmp_path, agents_src)

    res = team_init(ctl, str(tf), teams_root=str(teams_root))
    assert res["team"] == "myteam"
    # blueprints rendered per role x harness
    bps = ctl.list_blueprints("default", "default")
    assert sorted(bps) == ["myteam-dev-claude", "myteam-pm-claude"]
    bp = ctl.get_blueprint("default", "default", "myteam-dev-claude")
    assert bp.metadata.labels[api.LABEL_TEAM] == "myteam"
    # image capability selection: dev needs [go] -> kukeon.internal/go
    img = bp.spec.template["spec"]["containers"][0]["image"]
    assert img == "kukeon.internal/go"
    # catalog images registered in FROM order
    assert res["built"] == ["base", "go"]
    assert ctl.get_image("kukeon.internal/go")["spec"]["harness"] == "claude"
    # layered secrets: per-team overrides host-wide
    sec = ctl.get_secret("default", "default", "team-myteam")
    assert sec.spec.data == {"GLOBAL_KEY": "g1", "SHARED": "team"}
    # harness seeds written once, not overwritten
    seeded = teams_root / "myteam" / "state" / "claude" / "settings.json"
    assert seeded.read_text() == '{"seeded": true}'
    seeded.write_text("edited")
    team_init(ctl, str(tf), teams_root=str(teams_root))
    assert seeded.read_text() == "edited"


def test_team_prune_removes_dropped_roles(tmp_path, agents_src):
    ctl = Controller(str(tmp_path / "run"), runtime=FakeRuntime(),
                     gpu_devices=[])
    ctl.bootstrap()
    teams_root = tmp_path / "teams"
    tf = write_team_file(tmp_path, agents_src)
    team_init(ctl, str(tf), teams_root=str(teams_root))
    assert len(ctl.list_blueprints("default", "default")) == 2
    # drop the pm role -> its blueprint is pruned on re-init
    tf2 = write_team_file(tmp_path, agents_src,
                          roles="[{ref: dev, needs: {image: [go]}}]")
    res = team_init(ctl, str(tf2), teams_root=str(teams_root))
    assert "CellBlueprint/myteam-pm-claude" in res["pruned"]
    assert ctl.list_blueprints("default", "default") == ["myteam-dev-claude"]


def test_capability_mismatch_fails(tmp_path, agents_src):
    ctl = Controller(str(tmp_path / "run"), runtime=FakeRuntime(),
                     gpu_devices=[])
    ctl.bootstrap()
    tf = write_team_file(tmp_path, agents_src,
                         roles="[{ref: dev, needs: {image: [cuda]}}]")
    with pytest.raises(errors.ValidationError):
        team_init(ctl, str(tf), teams_root=str(tmp_path / "teams"))


def test_build_order_cycle_detection():
    from kukeon_amd.teams import CatalogImage, ImageCatalog
    cat = ImageCatalog(images=[CatalogImage(ref="a", base="b"),
                               CatalogImage(ref="b", base="a")])
    with pytest.raises(errors.ValidationError):
        build_order(cat)


def test_parse_rejects_wrong_apiversion():
    with pytest.raises(errors.ValidationError):
        parse_team_file("apiVersion: v1beta1\nkind: Role\n"
                        "metadata: {name: x}\n")
