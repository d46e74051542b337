module github.com/ashvardanian/stringzilla/golang

go 1.24
