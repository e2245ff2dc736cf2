module github.com/fast-ml/nezha-amd/go/nezha

go 1.21
