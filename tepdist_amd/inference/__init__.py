from tepdist_amd.inference.engine import Generator
