from gridllm_amd.worker.main import main

main()
