def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")
