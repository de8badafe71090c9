from .router import AdmissionChain, AdmissionError, AdmissionService, default_chain
