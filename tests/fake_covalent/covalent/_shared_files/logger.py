import logging

app_log = logging.getLogger("fake_covalent")
