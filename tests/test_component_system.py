"""Component system tests: resolver/loader with fake modules in sys.modules
(reference technique, tests/test_component_loader/*:21-46), config manager
load/save/update/default-file."""
import sys
import types

import pytest
import yaml

from detectmateservice_amd.components.base import CoreComponent, CoreConfig
from detectmateservice_amd.components.config_manager import ConfigManager
from detectmateservice_amd.components.loader import (
    ComponentLoader,
    ComponentLoadError,
    ConfigClassLoader,
)
from detectmateservice_amd.components.resolver import (
    ComponentResolutionError,
    ComponentResolver,
)


@pytest.fixture
def fake_library(monkeypatch):
    """Inject a fake component library package into sys.modules."""
    root = types.ModuleType("fakelib")
    root.__path__ = []  # looks like a package
    sub = types.ModuleType("fakelib.things")

    class FakeConfig(CoreConfig):
        knob: int = 1

    class FakeComponentConfig(CoreConfig):
        gain: float = 2.0

    class FakeComponent(CoreComponent):
        CONFIG_CLASS = FakeComponentConfig

        def process(self, data: bytes):
            return data[::-1]

    class NotAComponent:
        pass

    sub.FakeComponent = FakeComponent
    sub.FakeComponentConfig = FakeComponentConfig
    sub.FakeConfig = FakeConfig
    sub.NotAComponent = NotAComponent
    root.things = sub
    monkeypatch.setitem(sys.modules, "fakelib", root)
    monkeypatch.setitem(sys.modules, "fakelib.things", sub)
    return root


def test_resolver_finds_by_short_name(fake_library, monkeypatch):
    r = ComponentResolver(root_package="fakelib")
    # pkgutil.walk_packages won't find fake submodules (no real files), so
    # patch the walk result by resolving the dotted path directly:
    path, cfg = r.resolve("fakelib.things.FakeComponent")
    assert path == "fakelib.things.FakeComponent"
    assert cfg == "fakelib.things.FakeComponentConfig"


def test_resolver_real_library():
    r = ComponentResolver()
    path, cfg = r.resolve("NewValueDetector")
    assert path.endswith(".NewValueDetector")
    assert cfg.endswith(".NewValueDetectorConfig")


def test_resolver_unknown_raises():
    r = ComponentResolver()
    with pytest.raises(ComponentResolutionError):
        r.resolve("NoSuchComponentXYZ")


def test_loader_loads_and_typechecks(fake_library):
    loader = ComponentLoader(root_package="fakelib")
    comp = loader.load_component("fakelib.things.FakeComponent")
    assert comp.process(b"abc") == b"cba"
    with pytest.raises(ComponentLoadError):
        loader.load_component("fakelib.things.NotAComponent")
    with pytest.raises(ComponentLoadError):
        loader.load_component("fakelib.things.Missing")


def test_loader_passes_config(fake_library):
    loader = ComponentLoader(root_package="fakelib")
    comp = loader.load_component(
        "fakelib.things.FakeComponent", config={"gain": 5.0}
    )
    assert comp.config.gain == 5.0


def test_config_class_loader(fake_library):
    loader = ConfigClassLoader(base_package="fakelib")
    cls = loader.load_config_class("things.FakeConfig")
    assert cls.__name__ == "FakeConfig"
    cls2 = loader.load_config_class("fakelib.things.FakeConfig")
    assert cls2 is cls
    with pytest.raises(ComponentLoadError):
        loader.load_config_class("things.NotAComponent")


def test_config_manager_roundtrip(tmp_path):
    cfg_file = tmp_path / "component.yaml"
    cfg_file.write_text(
        yaml.safe_dump(
            {"detectors": {"NewValueDetector": {"data_use_training": 5}}}
        )
    )
    cm = ConfigManager(cfg_file)
    assert cm.component_section("NewValueDetector") == {"data_use_training": 5}
    cm.update({"detectors": {"NewValueDetector": {"data_use_training": 9}}})
    assert cm.component_section("NewValueDetector")["data_use_training"] == 9
    cm.save()
    reloaded = yaml.safe_load(cfg_file.read_text())
    assert reloaded["detectors"]["NewValueDetector"]["data_use_training"] == 9


def test_config_manager_creates_default_file(tmp_path):
    cfg_file = tmp_path / "sub" / "component.yaml"
    ConfigManager(cfg_file)
    assert cfg_file.exists()


def test_config_manager_rejects_non_mapping(tmp_path):
    cfg_file = tmp_path / "bad.yaml"
    cfg_file.write_text("- just\n- a list\n")
    with pytest.raises(Exception):
        ConfigManager(cfg_file)


def test_normalize_component_config():
    """params flattening + all_ prefix stripping (reference
    interfaces.md:74-82 config pipeline)."""
    from detectmateservice_amd.components.loader import normalize_component_config

    cfg = normalize_component_config({
        "params": {"threshold": 0.5, "seed": 1},
        "all_window": 8,
        "threshold": 0.9,  # explicit top-level wins over params
    })
    assert cfg["threshold"] == 0.9
    assert cfg["seed"] == 1
    assert cfg["window"] == 8
    assert cfg["all_window"] == 8  # original key kept
    assert normalize_component_config(None) is None
